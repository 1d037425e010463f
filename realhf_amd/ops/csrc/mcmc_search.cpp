// MCMC search over MFC -> (device mesh, dp/tp/pp) allocations (CPU).
// Reference semantics: csrc/search/{search.cpp, simulate.cpp, rpc.cpp} —
// multi-chain Metropolis search with beta annealing over per-MFC
// allocation choices, scoring candidates with a simulated DFG makespan
// that accounts for data dependencies, device occupancy and parameter-
// reallocation cost between strategies of the same role.
//
// Inputs are plain vectors prepared by realhf_amd/search/engine.py
// (per-MFC candidate lists with estimated compute seconds, memory bytes
// and mesh bitmasks over the node's 8 GPUs).
#include <torch/extension.h>

#include <algorithm>
#include <cstdint>
#include <random>
#include <vector>

namespace {

struct Candidate {
  uint32_t mesh;     // bitmask over GPUs
  double time_s;     // estimated MFC execution time
  double mem_bytes;  // persistent memory on each GPU of the mesh
  int strategy_id;   // identifies (dp, tp, pp) for realloc-cost lookup
};

struct Problem {
  int n_mfcs;
  int n_gpus;
  std::vector<std::vector<Candidate>> cands;  // per mfc
  std::vector<std::vector<int>> parents;      // dfg edges
  std::vector<int> role;                      // role id per mfc
  // realloc cost between two strategy ids of one role (seconds); square
  // matrix n_strategies x n_strategies, -1 => incompatible
  std::vector<double> realloc_cost;
  int n_strategies;
  double mem_cap_bytes;
};

double simulate(const Problem& P, const std::vector<int>& pick,
                double penalty_scale) {
  // per-GPU availability time
  std::vector<double> gpu_free(P.n_gpus, 0.0);
  std::vector<double> finish(P.n_mfcs, 0.0);
  std::vector<double> gpu_mem(P.n_gpus, 0.0);
  // memory: sum persistent mem of all MFCs' roles on each GPU (count one
  // allocation per (role, strategy) pair)
  std::vector<std::pair<int64_t, uint32_t>> seen;
  double mem_penalty = 0.0;
  for (int i = 0; i < P.n_mfcs; i++) {
    const Candidate& c = P.cands[i][pick[i]];
    int64_t key = ((int64_t)P.role[i] << 32) | c.strategy_id;
    bool dup = false;
    for (auto& s : seen)
      if (s.first == key) { dup = true; break; }
    if (!dup) {
      seen.push_back({key, c.mesh});
      for (int g = 0; g < P.n_gpus; g++)
        if (c.mesh & (1u << g)) gpu_mem[g] += c.mem_bytes;
    }
  }
  for (int g = 0; g < P.n_gpus; g++)
    if (gpu_mem[g] > P.mem_cap_bytes)
      mem_penalty += (gpu_mem[g] - P.mem_cap_bytes) / P.mem_cap_bytes;

  // makespan with realloc edges between same-role different-strategy MFCs
  for (int i = 0; i < P.n_mfcs; i++) {
    const Candidate& c = P.cands[i][pick[i]];
    double start = 0.0;
    for (int p : P.parents[i]) start = std::max(start, finish[p]);
    // realloc from the most recent same-role MFC with another strategy
    double extra = 0.0;
    for (int j = 0; j < i; j++) {
      if (P.role[j] != P.role[i]) continue;
      const Candidate& cj = P.cands[j][pick[j]];
      if (cj.strategy_id != c.strategy_id) {
        double rc = P.realloc_cost[(size_t)cj.strategy_id * P.n_strategies +
                                   c.strategy_id];
        if (rc < 0) return 1e18;  // incompatible
        extra = std::max(extra, rc);
      }
    }
    for (int g = 0; g < P.n_gpus; g++)
      if (c.mesh & (1u << g)) start = std::max(start, gpu_free[g]);
    double end = start + extra + c.time_s;
    finish[i] = end;
    for (int g = 0; g < P.n_gpus; g++)
      if (c.mesh & (1u << g)) gpu_free[g] = end;
  }
  double makespan = 0.0;
  for (double f : finish) makespan = std::max(makespan, f);
  return makespan * (1.0 + penalty_scale * mem_penalty);
}

}  // namespace

// returns (best_pick, best_cost)
std::pair<std::vector<int64_t>, double> mcmc_search(
    int64_t n_gpus, std::vector<std::vector<std::vector<double>>> cand_rows,
    std::vector<std::vector<int64_t>> parents_in,
    std::vector<int64_t> role_in, std::vector<double> realloc_cost_in,
    int64_t n_strategies, double mem_cap_bytes, int64_t n_chains,
    int64_t n_steps, int64_t seed) {
  Problem P;
  P.n_gpus = (int)n_gpus;
  P.n_mfcs = (int)cand_rows.size();
  P.n_strategies = (int)n_strategies;
  P.mem_cap_bytes = mem_cap_bytes;
  P.realloc_cost = realloc_cost_in;
  for (auto& r : role_in) P.role.push_back((int)r);
  for (auto& ps : parents_in) {
    std::vector<int> v;
    for (auto p : ps) v.push_back((int)p);
    P.parents.push_back(v);
  }
  for (auto& row : cand_rows) {
    std::vector<Candidate> cs;
    for (auto& c : row) {
      TORCH_CHECK(c.size() == 4);
      cs.push_back(Candidate{(uint32_t)(int64_t)c[0], c[1], c[2],
                             (int)(int64_t)c[3]});
    }
    TORCH_CHECK(!cs.empty(), "mfc with no feasible candidates");
    P.cands.push_back(cs);
  }

  std::vector<int> best;
  double best_cost = 1e30;
  for (int chain = 0; chain < n_chains; chain++) {
    std::mt19937 rng((unsigned)(seed + chain * 7919));
    std::vector<int> pick(P.n_mfcs);
    for (int i = 0; i < P.n_mfcs; i++)
      pick[i] = std::uniform_int_distribution<int>(
          0, (int)P.cands[i].size() - 1)(rng);
    double cost = simulate(P, pick, 10.0);
    std::vector<int> chain_best = pick;
    double chain_best_cost = cost;
    for (int step = 0; step < n_steps; step++) {
      double beta = 2.0 + 30.0 * step / std::max<int64_t>(1, n_steps);
      int i = std::uniform_int_distribution<int>(0, P.n_mfcs - 1)(rng);
      int old = pick[i];
      int alt = std::uniform_int_distribution<int>(
          0, (int)P.cands[i].size() - 1)(rng);
      if (alt == old) continue;
      pick[i] = alt;
      double c2 = simulate(P, pick, 10.0);
      double ref = std::max(cost, 1e-9);
      if (c2 <= cost ||
          std::uniform_real_distribution<double>(0, 1)(rng) <
              std::exp(-beta * (c2 - cost) / ref)) {
        cost = c2;
        if (cost < chain_best_cost) {
          chain_best_cost = cost;
          chain_best = pick;
        }
      } else {
        pick[i] = old;
      }
    }
    if (chain_best_cost < best_cost) {
      best_cost = chain_best_cost;
      best = chain_best;
    }
  }
  std::vector<int64_t> out;
  for (int p : best) out.push_back(p);
  return {out, best_cost};
}
