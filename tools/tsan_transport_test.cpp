// ThreadSanitizer harness for the C++ host engine (SURVEY.md section 5:
// the reference shipped no race detection; this framework provides a TSAN
// build target for the torch-free transport core: TcpMesh, CpuComm worker
// queue, PS frame machinery).
//
// Runs W "ranks" as in-process threads, each owning a CpuComm connected
// over loopback, and hammers sync + async collectives and the PS from
// multiple client threads.
//
// Build + run:  ./tools/run_tsan.sh
#include <atomic>
#include <cassert>
#include <cmath>
#include <cstdio>
#include <thread>
#include <vector>

#include "../torchmpi_amd/csrc/cpu_comm.h"
#include "../torchmpi_amd/csrc/ps.h"

using namespace tma;

int main() {
  const int W = 4;
  // Phase 0: create listeners, gather addresses.
  std::vector<std::unique_ptr<CpuComm>> comms;
  std::vector<std::string> addrs;
  for (int r = 0; r < W; ++r) {
    comms.push_back(std::make_unique<CpuComm>(r, "127.0.0.1"));
    addrs.push_back("127.0.0.1:" + std::to_string(comms[r]->listen_port()));
  }
  {
    std::vector<std::thread> ts;
    for (int r = 0; r < W; ++r)
      ts.emplace_back([&, r] { comms[r]->connect(r, addrs); });
    for (auto& t : ts) t.join();
  }

  fprintf(stderr, "phase0 done\n");
  // Phase 1: collectives from every rank thread, many async in flight.
  std::atomic<int> failures{0};
  {
    std::vector<std::thread> ts;
    for (int r = 0; r < W; ++r) {
      ts.emplace_back([&, r] {
        const size_t N = 4096;
        std::vector<std::vector<float>> bufs(16, std::vector<float>(N));
        for (int iter = 0; iter < 16; ++iter) {
          for (auto& b : bufs[iter % 16]) b = float(r + 1 + iter);
          auto w = comms[r]->enqueue([&, iter] {
            comms[r]->allreduce(bufs[iter % 16].data(), N, DType::F32,
                                RedOp::SUM);
          });
          w->wait();
          float expect = 0;
          for (int k = 0; k < W; ++k) expect += k + 1 + iter;
          if (std::fabs(bufs[iter % 16][0] - expect) > 1e-3) failures++;
        }
        fprintf(stderr, "rank %d allreduce loop done\n", r);
        // barrier storm
        for (int i = 0; i < 32; ++i)
          comms[r]->enqueue([&] { comms[r]->barrier(); })->wait();
      });
    }
    for (auto& t : ts) t.join();
  }

  fprintf(stderr, "phase1 done\n");
  // Phase 2: parameter server with concurrent client threads per rank.
  std::vector<std::unique_ptr<PsContext>> pss;
  std::vector<std::string> psaddrs;
  for (int r = 0; r < W; ++r) {
    pss.push_back(std::make_unique<PsContext>(r, "127.0.0.1"));
    psaddrs.push_back("127.0.0.1:" + std::to_string(pss[r]->listen_port()));
  }
  {
    std::vector<std::thread> ts;
    for (int r = 0; r < W; ++r)
      ts.emplace_back([&, r] { pss[r]->start(r, psaddrs); });
    for (auto& t : ts) t.join();
  }
  {
    const size_t N = 1000;
    std::vector<float> zeros(N, 0.f);
    for (int r = 0; r < W; ++r) pss[r]->register_tensor(7, N, DType::F32,
                                                        zeros.data());
    std::vector<std::thread> ts;
    for (int r = 0; r < W; ++r) {
      ts.emplace_back([&, r] {
        // two client threads per rank pushing adds concurrently
        std::thread c1([&] {
          std::vector<float> d(N, 1.0f);
          for (int i = 0; i < 8; ++i) pss[r]->send(7, d.data(), 1)->wait();
        });
        std::thread c2([&] {
          std::vector<float> d(N, 2.0f);
          for (int i = 0; i < 8; ++i) pss[r]->send(7, d.data(), 1)->wait();
        });
        c1.join();
        c2.join();
      });
    }
    for (auto& t : ts) t.join();
    // settle + verify: total added per rank = 8*1 + 8*2 = 24; W ranks => 96.
    // (enqueue ALL barriers before waiting any: barriers are collective.)
    std::vector<std::shared_ptr<Work>> bs;
    for (int r = 0; r < W; ++r)
      bs.push_back(comms[r]->enqueue([&, r] { comms[r]->barrier(); }));
    for (auto& b : bs) b->wait();
    std::vector<float> got(N);
    pss[0]->receive(7, got.data())->wait();
    if (std::fabs(got[0] - 24.f * W) > 1e-3) {
      printf("PS mismatch: %f vs %f\n", got[0], 24.f * W);
      failures++;
    }
    // Phase 2b: mutual large fetch — every rank receives the full tensor at
    // once while shards (4 MB each) far exceed the socket buffers. Drives
    // the loop's non-blocking out-queue drain + incremental frame reads
    // (the round-1 advisor deadlock, both directions) under TSAN.
    const size_t BIGN = (size_t)4 << 20;  // 16 MB fp32
    {
      std::vector<float> init(BIGN, 3.0f);
      for (int r = 0; r < W; ++r)
        pss[r]->register_tensor(8, BIGN, DType::F32, init.data());
      std::vector<std::thread> ts2;
      std::vector<int> bad(W, 0);
      for (int r = 0; r < W; ++r) {
        ts2.emplace_back([&, r] {
          std::vector<float> out(BIGN);
          pss[r]->receive(8, out.data())->wait();
          if (std::fabs(out[0] - 3.0f) > 1e-6 ||
              std::fabs(out[BIGN - 1] - 3.0f) > 1e-6)
            bad[r] = 1;
        });
      }
      for (auto& t : ts2) t.join();
      for (int r = 0; r < W; ++r)
        if (bad[r]) {
          printf("PS big fetch mismatch at rank %d\n", r);
          failures++;
        }
    }
    for (int r = 0; r < W; ++r) pss[r]->stop();
  }

  comms.clear();
  if (failures.load() == 0) {
    printf("TSAN harness OK (no functional failures)\n");
    return 0;
  }
  printf("FAILURES: %d\n", failures.load());
  return 1;
}
