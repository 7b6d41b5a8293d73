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
    (void)ndims;
    (void)dims;
    (void)periods;
    (void)reorder;
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
    // 1-D layout like shiftCartesianCoords
    for (int i = 0; i < maxdims; i++) {
        dims[i] = i == 0 ? world().getSize() : 1;
        periods[i] = 1;
        coords[i] = i == 0 ? thisRank() : 0;
    }
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
