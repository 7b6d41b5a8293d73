// MPI_* shim over MpiWorld (reference: the glue in
// tests/dist/mpi/mpi_native.cpp:59-774 — in Faasm it lives in the WASM
// host layer; here any C++ function linked against the runtime can use
// MPI directly). Implemented/stubbed boundary matches SURVEY Appendix A:
// stubs return an error and log, mirroring the reference's notImplemented.
#include "faabricamd/mpi/mpi.h"

#include "faabricamd/executor.h"
#include "faabricamd/mpi.h"
#include "faabricamd/util.h"

#include <cstring>

using namespace faabricamd;

// ------------------------- handle tables ------------------------------------

static faabric_datatype_t sInt8{ 0, 1 };
static faabric_datatype_t sInt16{ 1, 2 };
static faabric_datatype_t sInt32{ 2, 4 };
static faabric_datatype_t sInt64{ 3, 8 };
static faabric_datatype_t sUint8{ 4, 1 };
static faabric_datatype_t sUint16{ 5, 2 };
static faabric_datatype_t sUint32{ 6, 4 };
static faabric_datatype_t sUint64{ 7, 8 };
static faabric_datatype_t sLong{ 8, 8 };
static faabric_datatype_t sLongLong{ 9, 8 };
static faabric_datatype_t sFloat{ 10, 4 };
static faabric_datatype_t sDouble{ 11, 8 };
static faabric_datatype_t sChar{ 12, 1 };
static faabric_datatype_t sByte{ 13, 1 };
static faabric_datatype_t sNull{ 14, 1 };

MPI_Datatype MPI_INT8_T = &sInt8;
MPI_Datatype MPI_INT16_T = &sInt16;
MPI_Datatype MPI_INT32_T = &sInt32;
MPI_Datatype MPI_INT = &sInt32;
MPI_Datatype MPI_INT64_T = &sInt64;
MPI_Datatype MPI_UINT8_T = &sUint8;
MPI_Datatype MPI_UINT16_T = &sUint16;
MPI_Datatype MPI_UINT32_T = &sUint32;
MPI_Datatype MPI_UINT64_T = &sUint64;
MPI_Datatype MPI_LONG = &sLong;
MPI_Datatype MPI_LONG_LONG = &sLongLong;
MPI_Datatype MPI_LONG_LONG_INT = &sLongLong;
MPI_Datatype MPI_FLOAT = &sFloat;
MPI_Datatype MPI_DOUBLE = &sDouble;
MPI_Datatype MPI_CHAR = &sChar;
MPI_Datatype MPI_BYTE = &sByte;
MPI_Datatype MPI_DATATYPE_NULL = &sNull;

static faabric_op_t sOpMax{ 0 };
static faabric_op_t sOpMin{ 1 };
static faabric_op_t sOpSum{ 2 };
static faabric_op_t sOpProd{ 3 };
static faabric_op_t sOpLand{ 4 };
static faabric_op_t sOpLor{ 5 };
static faabric_op_t sOpBand{ 6 };
static faabric_op_t sOpBor{ 7 };
static faabric_op_t sOpMaxloc{ 8 };
static faabric_op_t sOpMinloc{ 9 };
static faabric_op_t sOpNull{ 10 };

MPI_Op MPI_MAX = &sOpMax;
MPI_Op MPI_MIN = &sOpMin;
MPI_Op MPI_SUM = &sOpSum;
MPI_Op MPI_PROD = &sOpProd;
MPI_Op MPI_LAND = &sOpLand;
MPI_Op MPI_LOR = &sOpLor;
MPI_Op MPI_BAND = &sOpBand;
MPI_Op MPI_BOR = &sOpBor;
MPI_Op MPI_MAXLOC = &sOpMaxloc;
MPI_Op MPI_MINLOC = &sOpMinloc;
MPI_Op MPI_OP_NULL = &sOpNull;

static faabric_communicator_t sCommWorld{ 0 };
MPI_Comm MPI_COMM_WORLD = &sCommWorld;

// ------------------------- helpers ------------------------------------------

static thread_local bool mpiInitialised = false;
static thread_local bool mpiFinalised = false;

static MpiWorld& world()
{
    return getMpiContext().getWorld();
}

static int thisRank()
{
    return getMpiContext().getRank();
}

static MpiDataType toType(MPI_Datatype t)
{
    switch (t->id) {
        case 2:
            return MpiDataType::INT32;
        case 3:
        case 8:
        case 9:
            return MpiDataType::INT64;
        case 7:
            return MpiDataType::UINT64;
        case 10:
            return MpiDataType::FLOAT;
        case 11:
            return MpiDataType::DOUBLE;
        default:
            return MpiDataType::BYTE;
    }
}

// For BYTE-mapped types the element count must scale by the type size
static int toCount(MPI_Datatype t, int count)
{
    return toType(t) == MpiDataType::BYTE ? count * t->size : count;
}

static MpiOp toOp(MPI_Op op)
{
    switch (op->id) {
        case 0:
            return MpiOp::MAX;
        case 1:
            return MpiOp::MIN;
        case 2:
            return MpiOp::SUM;
        case 3:
            return MpiOp::PROD;
        default:
            throw FaabricException(
              "MPI op not supported (matching the reference's "
              "max/min/sum/prod boundary)");
    }
}

#define NOT_IMPLEMENTED(name)                                                  \
    FAM_ERROR("%s is not implemented (stubbed like the reference)", name);     \
    return MPI_ERR_OTHER;

// ------------------------- lifecycle ----------------------------------------

int MPI_Init(int* argc, char*** argv)
{
    (void)argc;
    (void)argv;
    Message& msg = ExecutorContext::get().getMsg();
    auto& ctx = getMpiContext();
    if (msg.mpiRank == 0) {
        ctx.createWorld(msg);
    } else {
        ctx.joinWorld(msg);
    }
    mpiInitialised = true;
    mpiFinalised = false;
    return MPI_SUCCESS;
}

int MPI_Init_thread(int* argc, char*** argv, int required, int* provided)
{
    if (provided != nullptr) {
        *provided = MPI_THREAD_SINGLE;
    }
    (void)required;
    return MPI_Init(argc, argv);
}

int MPI_Initialized(int* flag)
{
    *flag = mpiInitialised ? 1 : 0;
    return MPI_SUCCESS;
}

int MPI_Finalize()
{
    mpiFinalised = true;
    return MPI_SUCCESS;
}

int MPI_Finalized(int* flag)
{
    *flag = mpiFinalised ? 1 : 0;
    return MPI_SUCCESS;
}

int MPI_Abort(MPI_Comm comm, int errorcode)
{
    (void)comm;
    FAM_ERROR("MPI_Abort called with code %d", errorcode);
    throw FaabricException("MPI_Abort");
}

int MPI_Query_thread(int* provided)
{
    *provided = MPI_THREAD_SINGLE;
    return MPI_SUCCESS;
}

// ------------------------- world info ---------------------------------------

int MPI_Comm_rank(MPI_Comm comm, int* rank)
{
    (void)comm;
    *rank = thisRank();
    return MPI_SUCCESS;
}

int MPI_Comm_size(MPI_Comm comm, int* size)
{
    (void)comm;
    *size = world().getSize();
    return MPI_SUCCESS;
}

int MPI_Get_processor_name(char* name, int* resultlen)
{
    const std::string& host = getSystemConfig().endpointHost;
    size_t n = std::min(host.size(), (size_t)MPI_MAX_PROCESSOR_NAME - 1);
    std::memcpy(name, host.data(), n);
    name[n] = '\0';
    *resultlen = (int)n;
    return MPI_SUCCESS;
}

int MPI_Get_version(int* version, int* subversion)
{
    (void)version;
    (void)subversion;
    NOT_IMPLEMENTED("MPI_Get_version");
}

double MPI_Wtime()
{
    return world().getWTime();
}

// ------------------------- point-to-point -----------------------------------

int MPI_Send(const void* buf,
             int count,
             MPI_Datatype datatype,
             int dest,
             int tag,
             MPI_Comm comm)
{
    (void)tag;
    (void)comm;
    world().send(thisRank(),
                 dest,
                 (const uint8_t*)buf,
                 toType(datatype),
                 toCount(datatype, count));
    return MPI_SUCCESS;
}

int MPI_Rsend(const void* buf,
              int count,
              MPI_Datatype datatype,
              int dest,
              int tag,
              MPI_Comm comm)
{
    (void)buf;
    (void)count;
    (void)datatype;
    (void)dest;
    (void)tag;
    (void)comm;
    NOT_IMPLEMENTED("MPI_Rsend");
}

int MPI_Recv(void* buf,
             int count,
             MPI_Datatype datatype,
             int source,
             int tag,
             MPI_Comm comm,
             MPI_Status* status)
{
    (void)tag;
    (void)comm;
    world().recv(source,
                 thisRank(),
                 (uint8_t*)buf,
                 toType(datatype),
                 toCount(datatype, count));
    if (status != MPI_STATUS_IGNORE) {
        status->MPI_SOURCE = source;
        status->MPI_ERROR = MPI_SUCCESS;
        status->bytesSize = count * datatype->size;
    }
    return MPI_SUCCESS;
}

int MPI_Sendrecv(const void* sendbuf,
                 int sendcount,
                 MPI_Datatype sendtype,
                 int dest,
                 int sendtag,
                 void* recvbuf,
                 int recvcount,
                 MPI_Datatype recvtype,
                 int source,
                 int recvtag,
                 MPI_Comm comm,
                 MPI_Status* status)
{
    (void)sendtag;
    (void)recvtag;
    (void)comm;
    (void)status;
    world().sendRecv((const uint8_t*)sendbuf,
                     toCount(sendtype, sendcount),
                     toType(sendtype),
                     dest,
                     (uint8_t*)recvbuf,
                     toCount(recvtype, recvcount),
                     toType(recvtype),
                     source,
                     thisRank());
    return MPI_SUCCESS;
}

int MPI_Isend(const void* buf,
              int count,
              MPI_Datatype datatype,
              int dest,
              int tag,
              MPI_Comm comm,
              MPI_Request* request)
{
    (void)tag;
    (void)comm;
    *request = world().isend(thisRank(),
                             dest,
                             (const uint8_t*)buf,
                             toType(datatype),
                             toCount(datatype, count));
    return MPI_SUCCESS;
}

int MPI_Irecv(void* buf,
              int count,
              MPI_Datatype datatype,
              int source,
              int tag,
              MPI_Comm comm,
              MPI_Request* request)
{
    (void)tag;
    (void)comm;
    *request = world().irecv(source,
                             thisRank(),
                             (uint8_t*)buf,
                             toType(datatype),
                             toCount(datatype, count));
    return MPI_SUCCESS;
}

int MPI_Wait(MPI_Request* request, MPI_Status* status)
{
    (void)status;
    world().awaitAsyncRequest(*request);
    return MPI_SUCCESS;
}

int MPI_Probe(int source, int tag, MPI_Comm comm, MPI_Status* status)
{
    (void)source;
    (void)tag;
    (void)comm;
    (void)status;
    NOT_IMPLEMENTED("MPI_Probe");
}

int MPI_Get_count(const MPI_Status* status,
                  MPI_Datatype datatype,
                  int* count)
{
    if (status->bytesSize % datatype->size != 0) {
        return MPI_ERR_OTHER;
    }
    *count = status->bytesSize / datatype->size;
    return MPI_SUCCESS;
}

// ------------------------- collectives --------------------------------------

int MPI_Barrier(MPI_Comm comm)
{
    (void)comm;
    world().barrier(thisRank());
    return MPI_SUCCESS;
}

int MPI_Bcast(void* buffer,
              int count,
              MPI_Datatype datatype,
              int root,
              MPI_Comm comm)
{
    (void)comm;
    world().broadcast(root,
                      thisRank(),
                      (uint8_t*)buffer,
                      toType(datatype),
                      toCount(datatype, count));
    return MPI_SUCCESS;
}

int MPI_Scatter(const void* sendbuf,
                int sendcount,
                MPI_Datatype sendtype,
                void* recvbuf,
                int recvcount,
                MPI_Datatype recvtype,
                int root,
                MPI_Comm comm)
{
    (void)recvcount;
    (void)recvtype;
    (void)comm;
    if (recvbuf == MPI_IN_PLACE && thisRank() == root) {
        // Root's chunk stays where it already is in sendbuf; MpiWorld's
        // scatter skips the aliased self-copy.
        recvbuf = (uint8_t*)sendbuf +
                  (size_t)root * (size_t)sendcount * sendtype->size;
    }
    world().scatter(root,
                    thisRank(),
                    (const uint8_t*)sendbuf,
                    (uint8_t*)recvbuf,
                    toType(sendtype),
                    toCount(sendtype, sendcount));
    return MPI_SUCCESS;
}

int MPI_Gather(const void* sendbuf,
               int sendcount,
               MPI_Datatype sendtype,
               void* recvbuf,
               int recvcount,
               MPI_Datatype recvtype,
               int root,
               MPI_Comm comm)
{
    (void)recvcount;
    (void)recvtype;
    (void)comm;
    if (sendbuf == MPI_IN_PLACE && thisRank() == root) {
        // Root's contribution is already in place in recvbuf; sendtype /
        // sendcount are ignored with MPI_IN_PLACE (MPI-2.2).
        sendbuf = (uint8_t*)recvbuf +
                  (size_t)root * (size_t)recvcount * recvtype->size;
        sendtype = recvtype;
        sendcount = recvcount;
    }
    world().gather(thisRank(),
                   root,
                   (const uint8_t*)sendbuf,
                   (uint8_t*)recvbuf,
                   toType(sendtype),
                   toCount(sendtype, sendcount));
    return MPI_SUCCESS;
}

int MPI_Allgather(const void* sendbuf,
                  int sendcount,
                  MPI_Datatype sendtype,
                  void* recvbuf,
                  int recvcount,
                  MPI_Datatype recvtype,
                  MPI_Comm comm)
{
    (void)recvcount;
    (void)recvtype;
    (void)comm;
    if (sendbuf == MPI_IN_PLACE) {
        sendbuf = (uint8_t*)recvbuf +
                  (size_t)thisRank() * (size_t)recvcount * recvtype->size;
        sendtype = recvtype;
        sendcount = recvcount;
    }
    world().allGather(thisRank(),
                      (const uint8_t*)sendbuf,
                      (uint8_t*)recvbuf,
                      toType(sendtype),
                      toCount(sendtype, sendcount));
    return MPI_SUCCESS;
}

int MPI_Reduce(const void* sendbuf,
               void* recvbuf,
               int count,
               MPI_Datatype datatype,
               MPI_Op op,
               int root,
               MPI_Comm comm)
{
    (void)comm;
    std::vector<uint8_t> inPlace;
    if (sendbuf == MPI_IN_PLACE && thisRank() == root) {
        // Input is taken from recvbuf; stage a copy so the reduction
        // never aliases its accumulation target.
        size_t bytes = (size_t)count * datatype->size;
        inPlace.assign((uint8_t*)recvbuf, (uint8_t*)recvbuf + bytes);
        sendbuf = inPlace.data();
    }
    world().reduce(thisRank(),
                   root,
                   (const uint8_t*)sendbuf,
                   (uint8_t*)recvbuf,
                   toType(datatype),
                   count,
                   toOp(op));
    return MPI_SUCCESS;
}

int MPI_Allreduce(const void* sendbuf,
                  void* recvbuf,
                  int count,
                  MPI_Datatype datatype,
                  MPI_Op op,
                  MPI_Comm comm)
{
    (void)comm;
    std::vector<uint8_t> inPlace;
    if (sendbuf == MPI_IN_PLACE) {
        size_t bytes = (size_t)count * datatype->size;
        inPlace.assign((uint8_t*)recvbuf, (uint8_t*)recvbuf + bytes);
        sendbuf = inPlace.data();
    }
    world().allReduce(thisRank(),
                      (const uint8_t*)sendbuf,
                      (uint8_t*)recvbuf,
                      toType(datatype),
                      count,
                      toOp(op));
    return MPI_SUCCESS;
}

int MPI_Reduce_scatter(const void* sendbuf,
                       void* recvbuf,
                       const int* recvcounts,
                       MPI_Datatype datatype,
                       MPI_Op op,
                       MPI_Comm comm)
{
    (void)comm;
    std::vector<uint8_t> inPlace;
    if (sendbuf == MPI_IN_PLACE) {
        // Full input lives in recvbuf (must hold all size*count elems)
        size_t bytes =
          (size_t)recvcounts[0] * world().getSize() * datatype->size;
        inPlace.assign((uint8_t*)recvbuf, (uint8_t*)recvbuf + bytes);
        sendbuf = inPlace.data();
    }
    // Uniform counts only (like RCCL's reduce-scatter)
    world().reduceScatter(thisRank(),
                          (const uint8_t*)sendbuf,
                          (uint8_t*)recvbuf,
                          toType(datatype),
                          recvcounts[0],
                          toOp(op));
    return MPI_SUCCESS;
}

int MPI_Scan(const void* sendbuf,
             void* recvbuf,
             int count,
             MPI_Datatype datatype,
             MPI_Op op,
             MPI_Comm comm)
{
    (void)comm;
    std::vector<uint8_t> inPlace;
    if (sendbuf == MPI_IN_PLACE) {
        size_t bytes = (size_t)count * datatype->size;
        inPlace.assign((uint8_t*)recvbuf, (uint8_t*)recvbuf + bytes);
        sendbuf = inPlace.data();
    }
    world().scan(thisRank(),
                 (const uint8_t*)sendbuf,
                 (uint8_t*)recvbuf,
                 toType(datatype),
                 count,
                 toOp(op));
    return MPI_SUCCESS;
}

int MPI_Alltoall(const void* sendbuf,
                 int sendcount,
                 MPI_Datatype sendtype,
                 void* recvbuf,
                 int recvcount,
                 MPI_Datatype recvtype,
                 MPI_Comm comm)
{
    (void)recvcount;
    (void)recvtype;
    (void)comm;
    std::vector<uint8_t> inPlace;
    if (sendbuf == MPI_IN_PLACE) {
        size_t bytes =
          (size_t)recvcount * world().getSize() * recvtype->size;
        inPlace.assign((uint8_t*)recvbuf, (uint8_t*)recvbuf + bytes);
        sendbuf = inPlace.data();
        sendtype = recvtype;
        sendcount = recvcount;
    }
    world().allToAll(thisRank(),
                     (const uint8_t*)sendbuf,
                     (uint8_t*)recvbuf,
                     toType(sendtype),
                     toCount(sendtype, sendcount));
    return MPI_SUCCESS;
}

// ------------------------- cartesian ----------------------------------------

int MPI_Cart_create(MPI_Comm old_comm,
                    int ndims,
                    const int* dims,
                    const int* periods,
                    int reorder,
                    MPI_Comm* comm)
{
    (void)periods;
    (void)reorder;
    world().setCartesianDims(ndims, dims);
    *comm = old_comm;
    return MPI_SUCCESS;
}

int MPI_Cart_rank(MPI_Comm comm, int* coords, int* rank)
{
    (void)comm;
    world().getRankFromCoords(rank, coords);
    return MPI_SUCCESS;
}

int MPI_Cart_get(MPI_Comm comm,
                 int maxdims,
                 int* dims,
                 int* periods,
                 int* coords)
{
    (void)comm;
    world().getCartesianGrid(thisRank(), maxdims, dims, periods, coords);
    return MPI_SUCCESS;
}

int MPI_Cart_shift(MPI_Comm comm,
                   int direction,
                   int disp,
                   int* rank_source,
                   int* rank_dest)
{
    (void)comm;
    world().shiftCartesianCoords(
      thisRank(), direction, disp, rank_source, rank_dest);
    return MPI_SUCCESS;
}

// ------------------------- types / memory -----------------------------------

int MPI_Type_size(MPI_Datatype type, int* size)
{
    *size = type->size;
    return MPI_SUCCESS;
}

int MPI_Type_contiguous(int count,
                        MPI_Datatype oldtype,
                        MPI_Datatype* newtype)
{
    auto* t = new faabric_datatype_t();
    t->id = 100;
    t->size = count * oldtype->size;
    *newtype = t;
    return MPI_SUCCESS;
}

int MPI_Type_commit(MPI_Datatype* type)
{
    (void)type;
    return MPI_SUCCESS;
}

int MPI_Type_free(MPI_Datatype* type)
{
    (void)type;
    NOT_IMPLEMENTED("MPI_Type_free");
}

int MPI_Alloc_mem(MPI_Aint size, MPI_Info info, void* baseptr)
{
    (void)info;
    *(void**)baseptr = malloc((size_t)size);
    return MPI_SUCCESS;
}

int MPI_Free_mem(void* base)
{
    free(base);
    return MPI_SUCCESS;
}

int MPI_Waitall(int count, MPI_Request* requests, MPI_Status* statuses)
{
    for (int i = 0; i < count; i++) {
        MPI_Wait(&requests[i],
                 statuses == MPI_STATUSES_IGNORE ? MPI_STATUS_IGNORE
                                                 : &statuses[i]);
    }
    return MPI_SUCCESS;
}

int MPI_Waitany(int count, MPI_Request* requests, int* index, MPI_Status* status)
{
    // Async requests complete in channel order here, so waiting on the
    // first outstanding request is a valid "any" (reference semantics:
    // recvBatchReturnLast drains in order)
    for (int i = 0; i < count; i++) {
        if (requests[i] >= 0) {
            MPI_Wait(&requests[i], status);
            requests[i] = -1;
            *index = i;
            return MPI_SUCCESS;
        }
    }
    *index = MPI_UNDEFINED;
    return MPI_SUCCESS;
}

int MPI_Request_free(MPI_Request* request)
{
    // Matching must stay aligned, so a freed request is drained rather
    // than abandoned
    if (request != nullptr && *request >= 0) {
        MPI_Wait(request, MPI_STATUS_IGNORE);
        *request = -1;
    }
    return MPI_SUCCESS;
}

int MPI_Gatherv(const void* sendbuf,
                int sendcount,
                MPI_Datatype sendtype,
                void* recvbuf,
                const int* recvcounts,
                const int* displs,
                MPI_Datatype recvtype,
                int root,
                MPI_Comm comm)
{
    (void)comm;
    int rank = thisRank();
    int size = world().getSize();
    if (rank == root) {
        for (int r = 0; r < size; r++) {
            uint8_t* dst =
              (uint8_t*)recvbuf + (size_t)displs[r] * recvtype->size;
            if (r == rank) {
                const uint8_t* src =
                  sendbuf == MPI_IN_PLACE ? dst : (const uint8_t*)sendbuf;
                if (src != dst) {
                    memcpy(dst, src, (size_t)recvcounts[r] * recvtype->size);
                }
            } else {
                world().recv(r, rank, dst, toType(recvtype),
                             toCount(recvtype, recvcounts[r]),
                             MpiMessageType::GATHER);
            }
        }
    } else {
        world().send(rank, root, (const uint8_t*)sendbuf, toType(sendtype),
                     toCount(sendtype, sendcount), MpiMessageType::GATHER);
    }
    return MPI_SUCCESS;
}

int MPI_Allgatherv(const void* sendbuf,
                   int sendcount,
                   MPI_Datatype sendtype,
                   void* recvbuf,
                   const int* recvcounts,
                   const int* displs,
                   MPI_Datatype recvtype,
                   MPI_Comm comm)
{
    int size = world().getSize();
    int rc = MPI_Gatherv(sendbuf, sendcount, sendtype, recvbuf, recvcounts,
                         displs, recvtype, 0, comm);
    if (rc != MPI_SUCCESS) {
        return rc;
    }
    // Broadcast the fully-gathered buffer (extent = max over ranks)
    size_t extent = 0;
    for (int r = 0; r < size; r++) {
        extent = std::max(
          extent, (size_t)displs[r] + (size_t)recvcounts[r]);
    }
    world().broadcast(0, thisRank(), (uint8_t*)recvbuf, toType(recvtype),
                      toCount(recvtype, (int)extent),
                      MpiMessageType::BROADCAST);
    return MPI_SUCCESS;
}

int MPI_Alltoallv(const void* sendbuf,
                  const int* sendcounts,
                  const int* sdispls,
                  MPI_Datatype sendtype,
                  void* recvbuf,
                  const int* recvcounts,
                  const int* rdispls,
                  MPI_Datatype recvtype,
                  MPI_Comm comm)
{
    (void)comm;
    int rank = thisRank();
    int size = world().getSize();
    // Post all irecvs, then send, then drain (pairwise exchange; the
    // reference's allToAll is also direct N^2 sends)
    std::vector<int> reqs;
    reqs.reserve(size);
    for (int r = 0; r < size; r++) {
        if (r == rank) {
            continue;
        }
        uint8_t* dst = (uint8_t*)recvbuf + (size_t)rdispls[r] * recvtype->size;
        reqs.push_back(world().irecv(r, rank, dst, toType(recvtype),
                                     toCount(recvtype, recvcounts[r]),
                                     MpiMessageType::ALLTOALL));
    }
    for (int r = 0; r < size; r++) {
        const uint8_t* src =
          (const uint8_t*)sendbuf + (size_t)sdispls[r] * sendtype->size;
        if (r == rank) {
            uint8_t* dst =
              (uint8_t*)recvbuf + (size_t)rdispls[r] * recvtype->size;
            memcpy(dst, src, (size_t)recvcounts[r] * recvtype->size);
            continue;
        }
        world().send(rank, r, src, toType(sendtype),
                     toCount(sendtype, sendcounts[r]),
                     MpiMessageType::ALLTOALL);
    }
    for (int req : reqs) {
        world().awaitAsyncRequest(req);
    }
    return MPI_SUCCESS;
}

int MPI_Comm_dup(MPI_Comm comm, MPI_Comm* newcomm)
{
    *newcomm = comm;
    return MPI_SUCCESS;
}

int MPI_Comm_free(MPI_Comm* comm)
{
    *comm = MPI_COMM_NULL;
    return MPI_SUCCESS;
}

int MPI_Comm_split(MPI_Comm comm, int color, int key, MPI_Comm* newcomm)
{
    (void)key;
    // Single-color split (the degenerate dup) is supported; true
    // sub-communicators are not part of the reference contract either
    if (color == 0) {
        *newcomm = comm;
        return MPI_SUCCESS;
    }
    NOT_IMPLEMENTED("MPI_Comm_split with color != 0");
}

int MPI_Comm_split_type(MPI_Comm comm,
                        int split_type,
                        int key,
                        MPI_Info info,
                        MPI_Comm* newcomm)
{
    (void)split_type;
    (void)key;
    (void)info;
    // Single node: every rank shares the node, so the split is a dup
    *newcomm = comm;
    return MPI_SUCCESS;
}

int MPI_Comm_create(MPI_Comm comm, MPI_Group group, MPI_Comm* newcomm)
{
    (void)comm;
    (void)group;
    (void)newcomm;
    NOT_IMPLEMENTED("MPI_Comm_create");
}

int MPI_Comm_create_group(MPI_Comm comm,
                          MPI_Group group,
                          int tag,
                          MPI_Comm* newcomm)
{
    (void)comm;
    (void)group;
    (void)tag;
    (void)newcomm;
    NOT_IMPLEMENTED("MPI_Comm_create_group");
}

int MPI_Comm_group(MPI_Comm comm, MPI_Group* group)
{
    (void)comm;
    *group = 0;
    return MPI_SUCCESS;
}

int MPI_Group_incl(MPI_Group group, int n, const int* ranks, MPI_Group* newgroup)
{
    (void)group;
    (void)n;
    (void)ranks;
    (void)newgroup;
    NOT_IMPLEMENTED("MPI_Group_incl");
}

int MPI_Group_free(MPI_Group* group)
{
    *group = MPI_GROUP_NULL;
    return MPI_SUCCESS;
}

int MPI_Op_create(void* user_fn, int commute, MPI_Op* op)
{
    (void)user_fn;
    (void)commute;
    (void)op;
    NOT_IMPLEMENTED("MPI_Op_create");
}

int MPI_Op_free(MPI_Op* op)
{
    *op = MPI_OP_NULL;
    return MPI_SUCCESS;
}

// One-sided RMA: header-only in the reference too (no MpiWorld
// implementation, reference src/mpi/MpiWorld.cpp) — reject loudly
int MPI_Win_create(void* base,
                   MPI_Aint size,
                   int disp_unit,
                   MPI_Info info,
                   MPI_Comm comm,
                   MPI_Win* win)
{
    (void)base; (void)size; (void)disp_unit; (void)info; (void)comm;
    (void)win;
    NOT_IMPLEMENTED("MPI_Win_create");
}

int MPI_Win_allocate_shared(MPI_Aint size,
                            int disp_unit,
                            MPI_Info info,
                            MPI_Comm comm,
                            void* baseptr,
                            MPI_Win* win)
{
    (void)size; (void)disp_unit; (void)info; (void)comm; (void)baseptr;
    (void)win;
    NOT_IMPLEMENTED("MPI_Win_allocate_shared");
}

int MPI_Win_shared_query(MPI_Win win,
                         int rank,
                         MPI_Aint* size,
                         int* disp_unit,
                         void* baseptr)
{
    (void)win; (void)rank; (void)size; (void)disp_unit; (void)baseptr;
    NOT_IMPLEMENTED("MPI_Win_shared_query");
}

int MPI_Win_get_attr(MPI_Win win,
                     int win_keyval,
                     void* attribute_val,
                     int* flag)
{
    (void)win; (void)win_keyval; (void)attribute_val; (void)flag;
    NOT_IMPLEMENTED("MPI_Win_get_attr");
}

int MPI_Win_fence(int assert_arg, MPI_Win win)
{
    (void)assert_arg;
    (void)win;
    NOT_IMPLEMENTED("MPI_Win_fence");
}

int MPI_Win_free(MPI_Win* win)
{
    (void)win;
    NOT_IMPLEMENTED("MPI_Win_free");
}

int MPI_Get(void* origin_addr,
            int origin_count,
            MPI_Datatype origin_datatype,
            int target_rank,
            MPI_Aint target_disp,
            int target_count,
            MPI_Datatype target_datatype,
            MPI_Win win)
{
    (void)origin_addr; (void)origin_count; (void)origin_datatype;
    (void)target_rank; (void)target_disp; (void)target_count;
    (void)target_datatype; (void)win;
    NOT_IMPLEMENTED("MPI_Get");
}

int MPI_Put(const void* origin_addr,
            int origin_count,
            MPI_Datatype origin_datatype,
            int target_rank,
            MPI_Aint target_disp,
            int target_count,
            MPI_Datatype target_datatype,
            MPI_Win win)
{
    (void)origin_addr; (void)origin_count; (void)origin_datatype;
    (void)target_rank; (void)target_disp; (void)target_count;
    (void)target_datatype; (void)win;
    NOT_IMPLEMENTED("MPI_Put");
}
