/*!
 * migbm Network — distributed collectives seam for the CPU learners.
 * Capability parity target: reference include/LightGBM/network.h + src/network/*.
 * MI355X-first design: the hot multi-GPU path uses RCCL directly inside the HIP learner
 * (see hip/), so the host Network only needs a correctness-grade backend. It is backed by
 * injected function pointers (parity with LGBM_NetworkInitWithFunctions) — the Python
 * package registers torch.distributed(gloo)-based callbacks for multi-process CPU
 * training/tests, replacing the reference's hand-rolled TCP socket mesh.
 */
#ifndef MIGBM_NETWORK_H_
#define MIGBM_NETWORK_H_

#include "common.h"

#include <cstring>
#include <vector>

namespace migbm {

/*! allgather callback: gathers `in_size` bytes from every rank into out (world*in_size,
 *  rank-major). Must be collective across all ranks. */
typedef void (*AllgatherFn)(const char* input, int input_size, char* output);

class Network {
 public:
  static void Init(int num_machines, int rank, AllgatherFn allgather);
  static void Free();
  static int num_machines() { return num_machines_; }
  static int rank() { return rank_; }
  static bool is_distributed() { return num_machines_ > 1; }

  /*! element-wise sum allreduce of doubles (histograms, stats). */
  static void AllreduceSum(double* data, size_t n);
  static void AllreduceSum(int64_t* data, size_t n);
  static void AllreduceSum(float* data, size_t n);
  /*! generic byte allgather: every rank contributes `size` bytes; out = world x size. */
  static void Allgather(const char* input, int size, char* output);
  /*! allgather with per-rank sizes */
  static void AllgatherV(const char* input, int my_size, const int* sizes, char* output);
  static double GlobalSyncUpByMean(double local);
  static double GlobalSyncUpBySum(double local);
  static int64_t GlobalSyncUpBySum(int64_t local);

 private:
  static int num_machines_;
  static int rank_;
  static AllgatherFn allgather_;
};

/*! standalone TCP collectives (socket_linker.cpp): full mesh + ring allgather,
 *  no external runtime required. machines = "ip:port,ip:port,...". */
void NetworkInitSockets(const std::string& machines, int local_listen_port,
                        int timeout_sec, int num_machines);
void NetworkFreeSockets();

}  // namespace migbm

#endif  // MIGBM_NETWORK_H_
