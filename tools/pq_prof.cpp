// host-decode profiler: times read_chunk per column across all row groups
// with a per-phase breakdown via env AURON_PARQUET_GPU toggles
#include <chrono>
#include <cstdio>
#include <thread>
#include <atomic>
#include <vector>
#include "parquet.h"
namespace auron { void pq_prof_dump(); }
using namespace auron;
using clk = std::chrono::steady_clock;
int main(int argc, char** argv) {
  ParquetFile f(argv[1]);
  int nth = argc > 2 ? atoi(argv[2]) : 1;
  auto t0 = clk::now();
  int G = f.num_row_groups(), C = (int)f.columns().size();
  std::vector<std::pair<int,int>> jobs;
  for (int g = 0; g < G; g++) for (int c = 0; c < C; c++) jobs.push_back({g,c});
  std::atomic<size_t> next{0};
  std::atomic<long> rows{0};
  std::vector<std::thread> ths;
  for (int t = 0; t < nth; t++) ths.emplace_back([&]{
    for (;;) { size_t j = next.fetch_add(1); if (j >= jobs.size()) break;
      auto cd = f.read_chunk(jobs[j].first, jobs[j].second);
      rows += cd.num_values; }
  });
  for (auto& t : ths) t.join();
  auron::pq_prof_dump();
  double s = std::chrono::duration<double>(clk::now() - t0).count();
  printf("threads=%d rows(2cols)=%ld time=%.2fs -> %.1f M values/s\n",
         nth, rows.load(), s, rows.load()/s/1e6);
}
