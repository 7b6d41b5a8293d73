// MPI compatibility header: the reference's custom mpi.h subset mapped
// onto faabricamd::MpiWorld (reference: include/faabric/mpi/mpi.h — ~76
// MPI_* symbols; implemented/stubbed boundary per SURVEY Appendix A).
// MPI programs written against this header run as faabric functions with
// one rank per GPU; collectives over HBM buffers go through RCCL/xGMI.
#pragma once

#include <cstddef>
#include <cstdint>

#define MPI_SUCCESS 0
#define MPI_ERR_OTHER 1

#define MPI_MAX_PROCESSOR_NAME 256
#define MPI_THREAD_SINGLE 0
#define MPI_THREAD_FUNNELED 1
#define MPI_THREAD_SERIALIZED 2
#define MPI_THREAD_MULTIPLE 3

// Opaque-ish handle types (tag + size for datatypes)
struct faabric_datatype_t
{
    int id;
    int size;
};

struct faabric_op_t
{
    int id;
};

struct faabric_communicator_t
{
    int id;
};

struct faabric_request_t
{
    int id;
};

struct faabric_info_t
{
    int id;
};

struct faabric_win_t
{
    int id;
};

typedef faabric_datatype_t* MPI_Datatype;
typedef faabric_op_t* MPI_Op;
typedef faabric_communicator_t* MPI_Comm;
typedef int MPI_Request;
typedef faabric_info_t* MPI_Info;
typedef faabric_win_t* MPI_Win;
typedef long MPI_Aint;
typedef int MPI_Fint;

struct MPI_Status
{
    int MPI_SOURCE;
    int MPI_TAG;
    int MPI_ERROR;
    int bytesSize;
};

typedef int MPI_Group;
#define MPI_GROUP_NULL (-1)
#define MPI_BOTTOM ((void*)0)
#define MPI_COMM_NULL ((MPI_Comm) nullptr)

#define MPI_STATUS_IGNORE ((MPI_Status*)nullptr)
#define MPI_STATUSES_IGNORE ((MPI_Status*)nullptr)
#define MPI_IN_PLACE ((void*)-1)
#define MPI_ANY_SOURCE -1
#define MPI_UNDEFINED -2

// Datatypes
extern MPI_Datatype MPI_INT8_T;
extern MPI_Datatype MPI_INT16_T;
extern MPI_Datatype MPI_INT32_T;
extern MPI_Datatype MPI_INT;
extern MPI_Datatype MPI_INT64_T;
extern MPI_Datatype MPI_UINT8_T;
extern MPI_Datatype MPI_UINT16_T;
extern MPI_Datatype MPI_UINT32_T;
extern MPI_Datatype MPI_UINT64_T;
extern MPI_Datatype MPI_LONG;
extern MPI_Datatype MPI_LONG_LONG;
extern MPI_Datatype MPI_LONG_LONG_INT;
extern MPI_Datatype MPI_FLOAT;
extern MPI_Datatype MPI_DOUBLE;
extern MPI_Datatype MPI_CHAR;
extern MPI_Datatype MPI_BYTE;
extern MPI_Datatype MPI_DATATYPE_NULL;

// Ops
extern MPI_Op MPI_MAX;
extern MPI_Op MPI_MIN;
extern MPI_Op MPI_SUM;
extern MPI_Op MPI_PROD;
extern MPI_Op MPI_LAND;
extern MPI_Op MPI_LOR;
extern MPI_Op MPI_BAND;
extern MPI_Op MPI_BOR;
extern MPI_Op MPI_MAXLOC;
extern MPI_Op MPI_MINLOC;
extern MPI_Op MPI_OP_NULL;

extern MPI_Comm MPI_COMM_WORLD;

// --- lifecycle ---
int MPI_Init(int* argc, char*** argv);
int MPI_Init_thread(int* argc, char*** argv, int required, int* provided);
int MPI_Initialized(int* flag);
int MPI_Finalize();
int MPI_Finalized(int* flag);
int MPI_Abort(MPI_Comm comm, int errorcode);
int MPI_Query_thread(int* provided);

// --- world info ---
int MPI_Comm_rank(MPI_Comm comm, int* rank);
int MPI_Comm_size(MPI_Comm comm, int* size);
int MPI_Get_processor_name(char* name, int* resultlen);
int MPI_Get_version(int* version, int* subversion);
double MPI_Wtime();

// --- point-to-point ---
int MPI_Send(const void* buf,
             int count,
             MPI_Datatype datatype,
             int dest,
             int tag,
             MPI_Comm comm);
int MPI_Rsend(const void* buf,
              int count,
              MPI_Datatype datatype,
              int dest,
              int tag,
              MPI_Comm comm);
int MPI_Recv(void* buf,
             int count,
             MPI_Datatype datatype,
             int source,
             int tag,
             MPI_Comm comm,
             MPI_Status* status);
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
                 MPI_Status* status);
int MPI_Isend(const void* buf,
              int count,
              MPI_Datatype datatype,
              int dest,
              int tag,
              MPI_Comm comm,
              MPI_Request* request);
int MPI_Irecv(void* buf,
              int count,
              MPI_Datatype datatype,
              int source,
              int tag,
              MPI_Comm comm,
              MPI_Request* request);
int MPI_Wait(MPI_Request* request, MPI_Status* status);
int MPI_Waitall(int count, MPI_Request* requests, MPI_Status* statuses);
int MPI_Waitany(int count,
                MPI_Request* requests,
                int* index,
                MPI_Status* status);
int MPI_Request_free(MPI_Request* request);
int MPI_Probe(int source, int tag, MPI_Comm comm, MPI_Status* status);
int MPI_Get_count(const MPI_Status* status,
                  MPI_Datatype datatype,
                  int* count);

// --- collectives ---
int MPI_Barrier(MPI_Comm comm);
int MPI_Bcast(void* buffer,
              int count,
              MPI_Datatype datatype,
              int root,
              MPI_Comm comm);
int MPI_Scatter(const void* sendbuf,
                int sendcount,
                MPI_Datatype sendtype,
                void* recvbuf,
                int recvcount,
                MPI_Datatype recvtype,
                int root,
                MPI_Comm comm);
int MPI_Gather(const void* sendbuf,
               int sendcount,
               MPI_Datatype sendtype,
               void* recvbuf,
               int recvcount,
               MPI_Datatype recvtype,
               int root,
               MPI_Comm comm);
int MPI_Allgather(const void* sendbuf,
                  int sendcount,
                  MPI_Datatype sendtype,
                  void* recvbuf,
                  int recvcount,
                  MPI_Datatype recvtype,
                  MPI_Comm comm);
int MPI_Reduce(const void* sendbuf,
               void* recvbuf,
               int count,
               MPI_Datatype datatype,
               MPI_Op op,
               int root,
               MPI_Comm comm);
int MPI_Allreduce(const void* sendbuf,
                  void* recvbuf,
                  int count,
                  MPI_Datatype datatype,
                  MPI_Op op,
                  MPI_Comm comm);
int MPI_Reduce_scatter(const void* sendbuf,
                       void* recvbuf,
                       const int* recvcounts,
                       MPI_Datatype datatype,
                       MPI_Op op,
                       MPI_Comm comm);
int MPI_Scan(const void* sendbuf,
             void* recvbuf,
             int count,
             MPI_Datatype datatype,
             MPI_Op op,
             MPI_Comm comm);
int MPI_Alltoall(const void* sendbuf,
                 int sendcount,
                 MPI_Datatype sendtype,
                 void* recvbuf,
                 int recvcount,
                 MPI_Datatype recvtype,
                 MPI_Comm comm);

int MPI_Gatherv(const void* sendbuf,
                int sendcount,
                MPI_Datatype sendtype,
                void* recvbuf,
                const int* recvcounts,
                const int* displs,
                MPI_Datatype recvtype,
                int root,
                MPI_Comm comm);
int MPI_Allgatherv(const void* sendbuf,
                   int sendcount,
                   MPI_Datatype sendtype,
                   void* recvbuf,
                   const int* recvcounts,
                   const int* displs,
                   MPI_Datatype recvtype,
                   MPI_Comm comm);
int MPI_Alltoallv(const void* sendbuf,
                  const int* sendcounts,
                  const int* sdispls,
                  MPI_Datatype sendtype,
                  void* recvbuf,
                  const int* recvcounts,
                  const int* rdispls,
                  MPI_Datatype recvtype,
                  MPI_Comm comm);

// --- communicator / group management ---
int MPI_Comm_dup(MPI_Comm comm, MPI_Comm* newcomm);
int MPI_Comm_free(MPI_Comm* comm);
int MPI_Comm_split(MPI_Comm comm, int color, int key, MPI_Comm* newcomm);
int MPI_Comm_split_type(MPI_Comm comm,
                        int split_type,
                        int key,
                        MPI_Info info,
                        MPI_Comm* newcomm);
int MPI_Comm_create(MPI_Comm comm, MPI_Group group, MPI_Comm* newcomm);
int MPI_Comm_create_group(MPI_Comm comm,
                          MPI_Group group,
                          int tag,
                          MPI_Comm* newcomm);
int MPI_Comm_group(MPI_Comm comm, MPI_Group* group);
int MPI_Group_incl(MPI_Group group,
                   int n,
                   const int* ranks,
                   MPI_Group* newgroup);
int MPI_Group_free(MPI_Group* group);
int MPI_Op_create(void* user_fn, int commute, MPI_Op* op);
int MPI_Op_free(MPI_Op* op);

// --- one-sided (RMA) — declared for API parity; the runtime rejects
// them at call time, matching the reference where RMA exists only in
// the header (reference include/faabric/mpi/mpi.h) ---
int MPI_Win_create(void* base,
                   MPI_Aint size,
                   int disp_unit,
                   MPI_Info info,
                   MPI_Comm comm,
                   MPI_Win* win);
int MPI_Win_allocate_shared(MPI_Aint size,
                            int disp_unit,
                            MPI_Info info,
                            MPI_Comm comm,
                            void* baseptr,
                            MPI_Win* win);
int MPI_Win_shared_query(MPI_Win win,
                         int rank,
                         MPI_Aint* size,
                         int* disp_unit,
                         void* baseptr);
int MPI_Win_get_attr(MPI_Win win,
                     int win_keyval,
                     void* attribute_val,
                     int* flag);
int MPI_Win_fence(int assert_arg, MPI_Win win);
int MPI_Win_free(MPI_Win* win);
int MPI_Get(void* origin_addr,
            int origin_count,
            MPI_Datatype origin_datatype,
            int target_rank,
            MPI_Aint target_disp,
            int target_count,
            MPI_Datatype target_datatype,
            MPI_Win win);
int MPI_Put(const void* origin_addr,
            int origin_count,
            MPI_Datatype origin_datatype,
            int target_rank,
            MPI_Aint target_disp,
            int target_count,
            MPI_Datatype target_datatype,
            MPI_Win win);

// --- cartesian topology ---
int MPI_Cart_create(MPI_Comm old_comm,
                    int ndims,
                    const int* dims,
                    const int* periods,
                    int reorder,
                    MPI_Comm* comm);
int MPI_Cart_rank(MPI_Comm comm, int* coords, int* rank);
int MPI_Cart_get(MPI_Comm comm,
                 int maxdims,
                 int* dims,
                 int* periods,
                 int* coords);
int MPI_Cart_shift(MPI_Comm comm,
                   int direction,
                   int disp,
                   int* rank_source,
                   int* rank_dest);

// --- types / memory ---
int MPI_Type_size(MPI_Datatype type, int* size);
int MPI_Type_contiguous(int count,
                        MPI_Datatype oldtype,
                        MPI_Datatype* newtype);
int MPI_Type_commit(MPI_Datatype* type);
int MPI_Type_free(MPI_Datatype* type);
int MPI_Alloc_mem(MPI_Aint size, MPI_Info info, void* baseptr);
int MPI_Free_mem(void* base);
