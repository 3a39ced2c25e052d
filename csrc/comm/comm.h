// Communication layer held by the library itself, not by Python.
//
// The reference's parallelism lives inside the library: pumipic::Library
// holds the MPI world comm (/root/reference/src/pumitally/
// PumiTallyImpl.cpp:238-241) and search(migrate) moves particles without
// the host app's involvement (:454).  This is the MI355X-native
// equivalent: one process per GPU, rank/world from torchrun-compatible
// environment variables (RANK, WORLD_SIZE, MASTER_ADDR, MASTER_PORT,
// LOCAL_RANK), RCCL over xGMI as the device data path.  A C++ host app
// linking pumitally::pumitally gets multi-GPU tallies with no Python and
// no MPI.
//
// Two transports behind one interface:
//   * TcpComm  -- plain sockets via the master address.  CPU fallback and
//     bootstrap; collectives are hub-and-spoke through rank 0.  Used for
//     the per-batch flux reduction in CPU runs, CI world-2 tests, and to
//     broadcast the RCCL unique id.  Never on a per-step critical path.
//   * RcclComm -- rcclComm per process over xGMI (7 p2p links x ~153 GB/s
//     per MI355X); device-buffer all-reduce for the flux tally and
//     send/recv-based all-to-all-v for partitioned particle records.
//     Bootstrap (unique-id exchange) rides a TcpComm.
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <utility>
#include <vector>

namespace pumitally {

class Comm {
public:
  virtual ~Comm() = default;
  virtual int rank() const = 0;
  virtual int world() const = 0;
  virtual void barrier() = 0;

  // In-place element-wise sum over ranks; host memory.
  virtual void allreduce_sum(double *data, int64_t n) = 0;
  virtual void allreduce_sum(int64_t *data, int64_t n) = 0;

  // In-place element-wise max over ranks; host memory (e.g. the slowest
  // rank's elapsed time in benchmarks).
  virtual void allreduce_max(double *data, int64_t n) = 0;

  // Root's buffer overwrites everyone's; host memory.
  virtual void bcast(void *data, int64_t bytes, int root) = 0;

  // One int64 per rank, returned indexed by rank; every rank gets all.
  virtual std::vector<int64_t> allgather(int64_t v) = 0;

  // Variable all-to-all of doubles: send holds concatenated
  // per-destination rows (send_counts[r] doubles bound for rank r, in
  // rank order); returns everything received, concatenated in source-rank
  // order.  Host memory.
  virtual std::vector<double> alltoallv(const double *send,
                                        const std::vector<int64_t> &send_counts) = 0;

  // Device-resident variants (RcclComm only; others throw).  d_data is
  // device memory on this rank's GPU; synchronous on return.
  // has_device_collectives() tells device-side callers whether to use
  // them directly or stage through host buffers.
  virtual bool has_device_collectives() const { return false; }
  virtual void allreduce_sum_device(double *d_data, int64_t n);
  // Device all-to-all-v: d_send as in alltoallv; recv_counts[r] doubles
  // are received from rank r into *d_recv (device buffer owned by the
  // comm, valid until the next device call); returns total received.
  virtual int64_t alltoallv_device(const double *d_send,
                                   const std::vector<int64_t> &send_counts,
                                   const std::vector<int64_t> &recv_counts,
                                   double **d_recv);
};

// TCP transport from the environment (RANK/WORLD_SIZE/MASTER_ADDR/
// MASTER_PORT; PUMITALLY_PORT overrides the comm port, which defaults to
// MASTER_PORT+371 so it never collides with a torchrun rendezvous on
// MASTER_PORT).  Throws if WORLD_SIZE>1 but the rendezvous fails.
std::unique_ptr<Comm> make_tcp_comm(int rank, int world,
                                    const std::string &addr, int port);

// RCCL transport on the given HIP device; unique id exchanged over a
// bootstrap TcpComm built from the same env.  Defined in the HIP TU;
// returns nullptr when no HIP device is available.
std::unique_ptr<Comm> make_rccl_comm(int rank, int world,
                                     const std::string &addr, int port,
                                     int device);

struct EnvComm {
  int rank = 0, world = 1, local_rank = 0;
  std::string addr = "127.0.0.1";
  int port = 0;
};
// Parse RANK/WORLD_SIZE/LOCAL_RANK/MASTER_ADDR/MASTER_PORT(+371 or
// PUMITALLY_PORT).  world==1 when unset.
EnvComm comm_env();

// The library's comm factory: nullptr when WORLD_SIZE<=1 (single
// process); otherwise RcclComm when want_gpu and a device exists
// (PUMITALLY_COMM=tcp forces TCP), else TcpComm.
std::unique_ptr<Comm> make_comm_from_env(bool want_gpu, int device);

} // namespace pumitally
