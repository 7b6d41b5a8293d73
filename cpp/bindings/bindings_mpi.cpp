// MPI world bindings. Functions running inside executors call mpi_init()
// then the collective ops; buffers can be Python bytes (host path) or raw
// device/host pointers (e.g. torch tensor.data_ptr()) for the RCCL path.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "faabricamd/executor.h"
#include "faabricamd/mpi.h"
#include "faabricamd/util.h"

namespace py = pybind11;
using namespace faabricamd;

namespace {

MpiWorld& world()
{
    return getMpiContext().getWorld();
}

} // namespace

void initMpiBindings(py::module_& m)
{
    py::enum_<MpiDataType>(m, "MpiDataType")
      .value("INT32", MpiDataType::INT32)
      .value("INT64", MpiDataType::INT64)
      .value("UINT64", MpiDataType::UINT64)
      .value("FLOAT", MpiDataType::FLOAT)
      .value("DOUBLE", MpiDataType::DOUBLE)
      .value("BYTE", MpiDataType::BYTE);

    py::enum_<MpiOp>(m, "MpiOp")
      .value("SUM", MpiOp::SUM)
      .value("MAX", MpiOp::MAX)
      .value("MIN", MpiOp::MIN)
      .value("PROD", MpiOp::PROD);

    // Initialise MPI for the currently-executing function: rank 0 creates
    // the world (gang-dispatching the other ranks), others join.
    // Returns (world_id, rank, world_size).
    m.def("mpi_init", [] {
        Message* msg = &ExecutorContext::get().getMsg();
        int worldId;
        int rank;
        int worldSize;
        {
            py::gil_scoped_release release;
            auto& ctx = getMpiContext();
            if (msg->mpiRank == 0 && !msg->isMpi) {
                throw FaabricException(
                  "mpi_init on message without isMpi set");
            }
            if (msg->mpiRank == 0) {
                ctx.createWorld(*msg);
            } else {
                ctx.joinWorld(*msg);
            }
            worldId = ctx.getWorldId();
            rank = ctx.getRank();
            worldSize = ctx.getWorld().getSize();
        }
        return py::make_tuple(worldId, rank, worldSize);
    });

    m.def("mpi_finalize", [] {
        py::gil_scoped_release release;
        auto& ctx = getMpiContext();
        if (ctx.getIsMpi()) {
            // Worlds are shared per host; clear when last local rank done
            // is handled by the registry clear on flush
        }
    });

    m.def("mpi_rank_of_current", [] {
        auto& ctx = getMpiContext();
        return py::make_tuple(ctx.getWorldId(), ctx.getRank());
    });

    m.def("mpi_barrier", [](int rank) {
        py::gil_scoped_release release;
        world().barrier(rank);
    });

    m.def("mpi_get_host_for_rank", [](int rank) {
        py::gil_scoped_release release;
        return world().getHostForRank(rank);
    });

    // ---------------- bytes-based host-path ops (tests, small payloads) ----
    m.def("mpi_send_bytes",
          [](int sendRank, int recvRank, const py::bytes& data) {
              std::string s = data;
              py::gil_scoped_release release;
              world().send(sendRank,
                           recvRank,
                           (const uint8_t*)s.data(),
                           MpiDataType::BYTE,
                           (int)s.size(),
                           MpiMessageType::NORMAL,
                           MpiBufferLoc::HOST);
          });
    m.def("mpi_recv_bytes", [](int sendRank, int recvRank, int count) {
        std::vector<uint8_t> buf(count);
        {
            py::gil_scoped_release release;
            world().recv(sendRank,
                         recvRank,
                         buf.data(),
                         MpiDataType::BYTE,
                         count,
                         MpiMessageType::NORMAL,
                         MpiBufferLoc::HOST);
        }
        return py::bytes((const char*)buf.data(), buf.size());
    });

    m.def("mpi_allreduce_bytes",
          [](int rank,
             const py::bytes& data,
             MpiDataType dtype,
             MpiOp op) {
              std::string s = data;
              int count = (int)(s.size() / mpiTypeSize(dtype));
              std::vector<uint8_t> out(s.size());
              {
                  py::gil_scoped_release release;
                  world().allReduce(rank,
                                    (const uint8_t*)s.data(),
                                    out.data(),
                                    dtype,
                                    count,
                                    op,
                                    MpiBufferLoc::HOST);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_reduce_bytes",
          [](int rank,
             int root,
             const py::bytes& data,
             MpiDataType dtype,
             MpiOp op) {
              std::string s = data;
              int count = (int)(s.size() / mpiTypeSize(dtype));
              std::vector<uint8_t> out(s.size());
              {
                  py::gil_scoped_release release;
                  world().reduce(rank,
                                 root,
                                 (const uint8_t*)s.data(),
                                 out.data(),
                                 dtype,
                                 count,
                                 op,
                                 MpiBufferLoc::HOST);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_bcast_bytes",
          [](int root, int rank, const py::bytes& data, int count) {
              std::string s = data;
              std::vector<uint8_t> buf(count);
              if (!s.empty()) {
                  std::memcpy(buf.data(),
                              s.data(),
                              std::min((size_t)count, s.size()));
              }
              {
                  py::gil_scoped_release release;
                  world().broadcast(root,
                                    rank,
                                    buf.data(),
                                    MpiDataType::BYTE,
                                    count,
                                    MpiMessageType::BROADCAST,
                                    MpiBufferLoc::HOST);
              }
              return py::bytes((const char*)buf.data(), buf.size());
          });

    m.def("mpi_scatter_bytes",
          [](int root, int rank, const py::bytes& sendData, int chunkBytes) {
              std::string s = sendData;
              std::vector<uint8_t> out(chunkBytes);
              {
                  py::gil_scoped_release release;
                  world().scatter(root,
                                  rank,
                                  (const uint8_t*)s.data(),
                                  out.data(),
                                  MpiDataType::BYTE,
                                  chunkBytes);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_gather_bytes",
          [](int rank, int root, const py::bytes& sendData, int worldSize) {
              std::string s = sendData;
              std::vector<uint8_t> out;
              if (rank == root) {
                  out.resize(s.size() * worldSize);
              }
              {
                  py::gil_scoped_release release;
                  world().gather(rank,
                                 root,
                                 (const uint8_t*)s.data(),
                                 out.data(),
                                 MpiDataType::BYTE,
                                 (int)s.size());
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_allgather_bytes",
          [](int rank, const py::bytes& sendData, int worldSize) {
              std::string s = sendData;
              std::vector<uint8_t> out(s.size() * worldSize);
              {
                  py::gil_scoped_release release;
                  world().allGather(rank,
                                    (const uint8_t*)s.data(),
                                    out.data(),
                                    MpiDataType::BYTE,
                                    (int)s.size(),
                                    MpiBufferLoc::HOST);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_alltoall_bytes",
          [](int rank, const py::bytes& sendData, int worldSize) {
              std::string s = sendData;
              int chunk = (int)(s.size() / worldSize);
              std::vector<uint8_t> out(s.size());
              {
                  py::gil_scoped_release release;
                  world().allToAll(rank,
                                   (const uint8_t*)s.data(),
                                   out.data(),
                                   MpiDataType::BYTE,
                                   chunk,
                                   MpiBufferLoc::HOST);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_scan_bytes",
          [](int rank, const py::bytes& data, MpiDataType dtype, MpiOp op) {
              std::string s = data;
              int count = (int)(s.size() / mpiTypeSize(dtype));
              std::vector<uint8_t> out(s.size());
              {
                  py::gil_scoped_release release;
                  world().scan(rank,
                               (const uint8_t*)s.data(),
                               out.data(),
                               dtype,
                               count,
                               op);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_reducescatter_bytes",
          [](int rank, const py::bytes& data, MpiDataType dtype,
             int recvCount, MpiOp op) {
              std::string s = data;
              std::vector<uint8_t> out((size_t)recvCount *
                                       mpiTypeSize(dtype));
              {
                  py::gil_scoped_release release;
                  world().reduceScatter(rank,
                                        (const uint8_t*)s.data(),
                                        out.data(),
                                        dtype,
                                        recvCount,
                                        op,
                                        MpiBufferLoc::HOST);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    m.def("mpi_sendrecv_bytes",
          [](int rank, int sendTo, int recvFrom, const py::bytes& data) {
              std::string s = data;
              std::vector<uint8_t> out(s.size());
              {
                  py::gil_scoped_release release;
                  world().sendRecv((const uint8_t*)s.data(),
                                   (int)s.size(),
                                   MpiDataType::BYTE,
                                   sendTo,
                                   out.data(),
                                   (int)s.size(),
                                   MpiDataType::BYTE,
                                   recvFrom,
                                   rank);
              }
              return py::bytes((const char*)out.data(), out.size());
          });

    // ---------------- pointer-based ops (device / RCCL path) ---------------
    // ptr args are raw addresses (e.g. torch tensor.data_ptr()); location
    // is probed via hipPointerGetAttributes
    m.def("mpi_allreduce_ptr",
          [](int rank,
             uintptr_t sendPtr,
             uintptr_t recvPtr,
             int count,
             MpiDataType dtype,
             MpiOp op) {
              py::gil_scoped_release release;
              world().allReduce(rank,
                                (const uint8_t*)sendPtr,
                                (uint8_t*)recvPtr,
                                dtype,
                                count,
                                op,
                                MpiBufferLoc::AUTO);
          });
    m.def("mpi_scan_ptr",
          [](int rank,
             uintptr_t sendPtr,
             uintptr_t recvPtr,
             int count,
             MpiDataType dtype,
             MpiOp op) {
              py::gil_scoped_release release;
              world().scan(rank,
                           (const uint8_t*)sendPtr,
                           (uint8_t*)recvPtr,
                           dtype,
                           count,
                           op,
                           MpiBufferLoc::AUTO);
          });
    m.def("mpi_reduce_ptr",
          [](int rank,
             int root,
             uintptr_t sendPtr,
             uintptr_t recvPtr,
             int count,
             MpiDataType dtype,
             MpiOp op) {
              py::gil_scoped_release release;
              world().reduce(rank,
                             root,
                             (const uint8_t*)sendPtr,
                             (uint8_t*)recvPtr,
                             dtype,
                             count,
                             op,
                             MpiBufferLoc::AUTO);
          });
    m.def("mpi_bcast_ptr",
          [](int root, int rank, uintptr_t ptr, int count,
             MpiDataType dtype) {
              py::gil_scoped_release release;
              world().broadcast(root,
                                rank,
                                (uint8_t*)ptr,
                                dtype,
                                count,
                                MpiMessageType::BROADCAST,
                                MpiBufferLoc::AUTO);
          });
    m.def("mpi_allgather_ptr",
          [](int rank,
             uintptr_t sendPtr,
             uintptr_t recvPtr,
             int count,
             MpiDataType dtype) {
              py::gil_scoped_release release;
              world().allGather(rank,
                                (const uint8_t*)sendPtr,
                                (uint8_t*)recvPtr,
                                dtype,
                                count,
                                MpiBufferLoc::AUTO);
          });
    m.def("mpi_reducescatter_ptr",
          [](int rank,
             uintptr_t sendPtr,
             uintptr_t recvPtr,
             int recvCount,
             MpiDataType dtype,
             MpiOp op) {
              py::gil_scoped_release release;
              world().reduceScatter(rank,
                                    (const uint8_t*)sendPtr,
                                    (uint8_t*)recvPtr,
                                    dtype,
                                    recvCount,
                                    op,
                                    MpiBufferLoc::AUTO);
          });
    m.def("mpi_alltoall_ptr",
          [](int rank,
             uintptr_t sendPtr,
             uintptr_t recvPtr,
             int countPerRank,
             MpiDataType dtype) {
              py::gil_scoped_release release;
              world().allToAll(rank,
                               (const uint8_t*)sendPtr,
                               (uint8_t*)recvPtr,
                               dtype,
                               countPerRank,
                               MpiBufferLoc::AUTO);
          });
    m.def("mpi_send_ptr",
          [](int sendRank,
             int recvRank,
             uintptr_t ptr,
             int count,
             MpiDataType dtype) {
              py::gil_scoped_release release;
              world().send(sendRank,
                           recvRank,
                           (const uint8_t*)ptr,
                           dtype,
                           count,
                           MpiMessageType::NORMAL,
                           MpiBufferLoc::AUTO);
          });
    m.def("mpi_recv_ptr",
          [](int sendRank,
             int recvRank,
             uintptr_t ptr,
             int count,
             MpiDataType dtype) {
              py::gil_scoped_release release;
              world().recv(sendRank,
                           recvRank,
                           (uint8_t*)ptr,
                           dtype,
                           count,
                           MpiMessageType::NORMAL,
                           MpiBufferLoc::AUTO);
          });

    m.def("mpi_isend_ptr",
          [](int sendRank,
             int recvRank,
             uintptr_t ptr,
             int count,
             MpiDataType dtype) {
              py::gil_scoped_release release;
              return world().isend(sendRank,
                                   recvRank,
                                   (const uint8_t*)ptr,
                                   dtype,
                                   count,
                                   MpiMessageType::NORMAL,
                                   MpiBufferLoc::AUTO);
          });
    m.def("mpi_irecv_ptr",
          [](int sendRank,
             int recvRank,
             uintptr_t ptr,
             int count,
             MpiDataType dtype) {
              py::gil_scoped_release release;
              return world().irecv(sendRank,
                                   recvRank,
                                   (uint8_t*)ptr,
                                   dtype,
                                   count,
                                   MpiMessageType::NORMAL,
                                   MpiBufferLoc::AUTO);
          });
    m.def("mpi_await", [](int requestId) {
        py::gil_scoped_release release;
        world().awaitAsyncRequest(requestId);
    });

    m.def("mpi_msg_count_details", [](int rank) {
        return world().getMsgCountDetails(rank);
    });
    m.def("mpi_clear_registry", [] { MpiWorldRegistry::get().clear(); });
}
