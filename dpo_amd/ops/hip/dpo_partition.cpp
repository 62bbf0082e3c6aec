// Native multi-level k-way graph partitioner (CPU, C++).
//
// Replaces the reference's offline KaHIP runs (reference graph/<k>/
// <preset>/ partition files were produced out-of-repo)
// with an in-framework partitioner: heavy-edge-matching coarsening,
// greedy graph-growing initial partition, and boundary
// Fiduccia-Mattheyses refinement with hill climbing + rollback at every
// uncoarsening level, plus a connected-components fixup (RBCD converges
// poorly when an agent's subgraph is disconnected). Multi-restart keeps
// the best cut. Deterministic for a fixed seed.
//
// Exposed via C ABI for ctypes: dpo_partition_multilevel().

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <numeric>
#include <queue>
#include <random>
#include <vector>

namespace {

struct Graph {
  int n = 0;
  std::vector<int> xadj;    // CSR offsets (n+1)
  std::vector<int> adjncy;  // neighbor ids
  std::vector<double> adjw; // edge weights
  std::vector<double> vwgt; // vertex weights
};

struct Level {
  Graph g;
  std::vector<int> cmap;  // fine vertex -> coarse vertex
};

Graph coarsen(const Graph& g, std::vector<int>& cmap, std::mt19937& rng) {
  const int n = g.n;
  std::vector<int> match(n, -1);
  std::vector<int> order(n);
  std::iota(order.begin(), order.end(), 0);
  std::shuffle(order.begin(), order.end(), rng);
  for (int u : order) {
    if (match[u] >= 0) continue;
    int best = -1;
    double bw = -1.0;
    for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e) {
      int v = g.adjncy[e];
      if (match[v] < 0 && v != u && g.adjw[e] > bw) {
        bw = g.adjw[e];
        best = v;
      }
    }
    match[u] = (best >= 0) ? best : u;
    if (best >= 0) match[best] = u;
  }
  cmap.assign(n, -1);
  int nc = 0;
  for (int u = 0; u < n; ++u)
    if (cmap[u] < 0) {
      cmap[u] = nc;
      cmap[match[u]] = nc;
      ++nc;
    }
  // build coarse graph
  Graph c;
  c.n = nc;
  c.vwgt.assign(nc, 0.0);
  for (int u = 0; u < n; ++u) c.vwgt[cmap[u]] += g.vwgt[u];
  std::vector<std::vector<std::pair<int, double>>> tmp(nc);
  std::vector<int> seen(nc, -1);
  std::vector<int> pos(nc, 0);
  for (int u = 0; u < n; ++u) {
    int cu = cmap[u];
    for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e) {
      int cv = cmap[g.adjncy[e]];
      if (cu == cv) continue;
      if (seen[cv] == cu) {
        tmp[cu][pos[cv]].second += g.adjw[e];
      } else {
        seen[cv] = cu;
        pos[cv] = (int)tmp[cu].size();
        tmp[cu].push_back({cv, g.adjw[e]});
      }
    }
  }
  c.xadj.assign(nc + 1, 0);
  for (int u = 0; u < nc; ++u) c.xadj[u + 1] = c.xadj[u] + (int)tmp[u].size();
  c.adjncy.resize(c.xadj[nc]);
  c.adjw.resize(c.xadj[nc]);
  for (int u = 0; u < nc; ++u)
    for (size_t j = 0; j < tmp[u].size(); ++j) {
      c.adjncy[c.xadj[u] + j] = tmp[u][j].first;
      c.adjw[c.xadj[u] + j] = tmp[u][j].second;
    }
  return c;
}

double cut_of(const Graph& g, const std::vector<int>& part) {
  double cut = 0.0;
  for (int u = 0; u < g.n; ++u)
    for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e)
      if (part[u] != part[g.adjncy[e]]) cut += g.adjw[e];
  return cut / 2.0;
}

void initial_partition(const Graph& g, int k, double max_w,
                       std::vector<int>& part, std::mt19937& rng) {
  const int n = g.n;
  part.assign(n, -1);
  std::vector<double> wgt(k, 0.0);
  // spread seeds by BFS farthest-point
  std::vector<int> seeds;
  std::uniform_int_distribution<int> uni(0, n - 1);
  seeds.push_back(uni(rng));
  std::vector<int> dist(n);
  for (int s = 1; s < k; ++s) {
    std::fill(dist.begin(), dist.end(), -1);
    std::queue<int> q;
    for (int x : seeds) {
      dist[x] = 0;
      q.push(x);
    }
    while (!q.empty()) {
      int u = q.front();
      q.pop();
      for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e) {
        int v = g.adjncy[e];
        if (dist[v] < 0) {
          dist[v] = dist[u] + 1;
          q.push(v);
        }
      }
    }
    int far = 0, fd = -1;
    for (int u = 0; u < n; ++u) {
      int du = dist[u] < 0 ? 1 << 28 : dist[u];
      if (du > fd) {
        fd = du;
        far = u;
      }
    }
    seeds.push_back(far);
  }
  // grow regions, strongest-connection first, lightest region priority
  using HE = std::pair<double, int>;
  std::vector<std::priority_queue<HE>> heaps(k);
  for (int p = 0; p < k; ++p) {
    part[seeds[p]] = p;
    wgt[p] += g.vwgt[seeds[p]];
    for (int e = g.xadj[seeds[p]]; e < g.xadj[seeds[p] + 1]; ++e)
      heaps[p].push({g.adjw[e], g.adjncy[e]});
  }
  bool active = true;
  while (active) {
    active = false;
    // grow the lightest region one step
    std::vector<int> order(k);
    std::iota(order.begin(), order.end(), 0);
    std::sort(order.begin(), order.end(),
              [&](int a, int b) { return wgt[a] < wgt[b]; });
    for (int p : order) {
      auto& h = heaps[p];
      while (!h.empty()) {
        auto [w, v] = h.top();
        h.pop();
        if (part[v] >= 0) continue;
        if (wgt[p] + g.vwgt[v] > max_w) break;
        part[v] = p;
        wgt[p] += g.vwgt[v];
        for (int e = g.xadj[v]; e < g.xadj[v + 1]; ++e)
          if (part[g.adjncy[e]] < 0)
            h.push({g.adjw[e], g.adjncy[e]});
        active = true;
        break;
      }
    }
  }
  for (int u = 0; u < n; ++u)
    if (part[u] < 0) {
      int p = (int)(std::min_element(wgt.begin(), wgt.end()) - wgt.begin());
      part[u] = p;
      wgt[p] += g.vwgt[u];
    }
}

// One FM pass with hill climbing: tentatively apply best moves (any
// gain) with balance constraint, lock vertices, remember the best
// prefix, roll back the tail. Returns cut improvement.
double fm_pass(const Graph& g, int k, double max_w, std::vector<int>& part,
               int max_moves) {
  const int n = g.n;
  std::vector<double> wgt(k, 0.0);
  for (int u = 0; u < n; ++u) wgt[part[u]] += g.vwgt[u];

  // connection weights of u to each partition (sparse recompute)
  auto conn_to = [&](int u, std::vector<double>& conn) {
    conn.assign(k, 0.0);
    for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e)
      conn[part[g.adjncy[e]]] += g.adjw[e];
  };

  using HE = std::tuple<double, int, int>;  // gain, vertex, target
  std::priority_queue<HE> heap;
  std::vector<double> conn;
  auto push_vertex = [&](int u) {
    conn_to(u, conn);
    int pu = part[u];
    for (int p = 0; p < k; ++p) {
      if (p == pu || conn[p] <= 0.0) continue;
      heap.push({conn[p] - conn[pu], u, p});
    }
  };
  for (int u = 0; u < n; ++u) {
    bool boundary = false;
    for (int e = g.xadj[u]; e < g.xadj[u + 1] && !boundary; ++e)
      boundary = part[g.adjncy[e]] != part[u];
    if (boundary) push_vertex(u);
  }

  std::vector<char> locked(n, 0);
  std::vector<std::pair<int, int>> moves;  // (vertex, old part)
  double cum_gain = 0.0, best_gain = 0.0;
  int best_len = 0;
  int stall = 0;
  while (!heap.empty() && (int)moves.size() < max_moves && stall < 200) {
    auto [gain, u, tp] = heap.top();
    heap.pop();
    if (locked[u] || part[u] == tp) continue;
    conn_to(u, conn);
    double real_gain = conn[tp] - conn[part[u]];
    if (real_gain < gain - 1e-12) {  // stale entry: reinsert fresh
      if (conn[tp] > 0) heap.push({real_gain, u, tp});
      continue;
    }
    if (wgt[tp] + g.vwgt[u] > max_w) continue;
    // apply
    int pu = part[u];
    part[u] = tp;
    wgt[pu] -= g.vwgt[u];
    wgt[tp] += g.vwgt[u];
    locked[u] = 1;
    moves.push_back({u, pu});
    cum_gain += real_gain;
    if (cum_gain > best_gain + 1e-12) {
      best_gain = cum_gain;
      best_len = (int)moves.size();
      stall = 0;
    } else {
      ++stall;
    }
    for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e) {
      int v = g.adjncy[e];
      if (!locked[v]) push_vertex(v);
    }
  }
  // roll back past the best prefix
  for (int i = (int)moves.size() - 1; i >= best_len; --i) {
    auto [u, pu] = moves[i];
    part[u] = pu;
  }
  return best_gain;
}

void refine(const Graph& g, int k, double max_w, std::vector<int>& part) {
  for (int pass = 0; pass < 12; ++pass) {
    double gain = fm_pass(g, k, max_w, part, g.n);
    if (gain <= 1e-12) break;
  }
}

// Move every non-largest connected component of each class to the
// neighboring class it is most strongly connected to.
void connectivity_fixup(const Graph& g, int k, double max_w,
                        std::vector<int>& part) {
  const int n = g.n;
  for (int iter = 0; iter < 3; ++iter) {
    std::vector<double> wgt(k, 0.0);
    for (int u = 0; u < n; ++u) wgt[part[u]] += g.vwgt[u];
    std::vector<int> comp(n, -1);
    int nc = 0;
    std::vector<std::vector<int>> comp_members;
    std::vector<double> comp_w;
    for (int s = 0; s < n; ++s) {
      if (comp[s] >= 0) continue;
      comp[s] = nc;
      std::vector<int> stack{s}, members{s};
      double w = g.vwgt[s];
      while (!stack.empty()) {
        int u = stack.back();
        stack.pop_back();
        for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e) {
          int v = g.adjncy[e];
          if (comp[v] < 0 && part[v] == part[s]) {
            comp[v] = nc;
            stack.push_back(v);
            members.push_back(v);
            w += g.vwgt[v];
          }
        }
      }
      comp_members.push_back(std::move(members));
      comp_w.push_back(w);
      ++nc;
    }
    // largest component per class stays
    std::vector<int> largest(k, -1);
    for (int c = 0; c < nc; ++c) {
      int p = part[comp_members[c][0]];
      if (largest[p] < 0 || comp_w[c] > comp_w[largest[p]])
        largest[p] = c;
    }
    bool moved = false;
    for (int c = 0; c < nc; ++c) {
      int p = part[comp_members[c][0]];
      if (c == largest[p]) continue;
      // strongest-connected neighbor class
      std::vector<double> conn(k, 0.0);
      for (int u : comp_members[c])
        for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e)
          if (part[g.adjncy[e]] != p) conn[part[g.adjncy[e]]] += g.adjw[e];
      int best = -1;
      double bw = 0.0;
      for (int q = 0; q < k; ++q)
        if (conn[q] > bw) {
          bw = conn[q];
          best = q;
        }
      if (best >= 0 && wgt[best] + comp_w[c] <= max_w * 1.15) {
        for (int u : comp_members[c]) part[u] = best;
        wgt[best] += comp_w[c];
        wgt[p] -= comp_w[c];
        moved = true;
      }
    }
    if (!moved) break;
  }
}

// Coarsening that only matches vertices within the same class, so an
// existing partition projects losslessly onto the coarse graph
// (iterated-multilevel V-cycles, the main quality lever of modern
// multilevel partitioners).
Graph coarsen_respecting(const Graph& g, const std::vector<int>& part,
                         std::vector<int>& cmap, std::mt19937& rng) {
  const int n = g.n;
  std::vector<int> match(n, -1);
  std::vector<int> order(n);
  std::iota(order.begin(), order.end(), 0);
  std::shuffle(order.begin(), order.end(), rng);
  for (int u : order) {
    if (match[u] >= 0) continue;
    int best = -1;
    double bw = -1.0;
    for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e) {
      int v = g.adjncy[e];
      if (match[v] < 0 && v != u && part[v] == part[u] && g.adjw[e] > bw) {
        bw = g.adjw[e];
        best = v;
      }
    }
    match[u] = (best >= 0) ? best : u;
    if (best >= 0) match[best] = u;
  }
  cmap.assign(n, -1);
  int nc = 0;
  for (int u = 0; u < n; ++u)
    if (cmap[u] < 0) {
      cmap[u] = nc;
      cmap[match[u]] = nc;
      ++nc;
    }
  Graph c;
  c.n = nc;
  c.vwgt.assign(nc, 0.0);
  for (int u = 0; u < n; ++u) c.vwgt[cmap[u]] += g.vwgt[u];
  std::vector<std::vector<std::pair<int, double>>> tmp(nc);
  std::vector<int> seen(nc, -1);
  std::vector<int> pos(nc, 0);
  for (int u = 0; u < n; ++u) {
    int cu = cmap[u];
    for (int e = g.xadj[u]; e < g.xadj[u + 1]; ++e) {
      int cv = cmap[g.adjncy[e]];
      if (cu == cv) continue;
      if (seen[cv] == cu) {
        tmp[cu][pos[cv]].second += g.adjw[e];
      } else {
        seen[cv] = cu;
        pos[cv] = (int)tmp[cu].size();
        tmp[cu].push_back({cv, g.adjw[e]});
      }
    }
  }
  c.xadj.assign(nc + 1, 0);
  for (int u = 0; u < nc; ++u) c.xadj[u + 1] = c.xadj[u] + (int)tmp[u].size();
  c.adjncy.resize(c.xadj[nc]);
  c.adjw.resize(c.xadj[nc]);
  for (int u = 0; u < nc; ++u)
    for (size_t j = 0; j < tmp[u].size(); ++j) {
      c.adjncy[c.xadj[u] + j] = tmp[u][j].first;
      c.adjw[c.xadj[u] + j] = tmp[u][j].second;
    }
  return c;
}

// One V-cycle: coarsen respecting `part`, refine coarse-to-fine.
std::vector<int> vcycle(const Graph& g0, int k, double max_w,
                        std::vector<int> part, int target,
                        std::mt19937& rng) {
  std::vector<Level> levels;
  Graph g = g0;
  std::vector<int> cur = std::move(part);
  while (g.n > target) {
    Level lv;
    Graph c = coarsen_respecting(g, cur, lv.cmap, rng);
    if (c.n >= (int)(g.n * 0.95)) break;
    std::vector<int> cpart(c.n);
    for (int u = 0; u < g.n; ++u) cpart[lv.cmap[u]] = cur[u];
    lv.g = std::move(g);
    g = std::move(c);
    cur = std::move(cpart);
    levels.push_back(std::move(lv));
  }
  refine(g, k, max_w, cur);
  while (!levels.empty()) {
    Level lv = std::move(levels.back());
    levels.pop_back();
    std::vector<int> fine(lv.g.n);
    for (int u = 0; u < lv.g.n; ++u) fine[u] = cur[lv.cmap[u]];
    cur = std::move(fine);
    refine(lv.g, k, max_w, cur);
    g = std::move(lv.g);
  }
  return cur;
}

std::vector<int> multilevel_once(const Graph& g0, int k, double imbalance,
                                 unsigned seed) {
  std::mt19937 rng(seed);
  double total = 0.0;
  for (double w : g0.vwgt) total += w;
  const double max_w = (1.0 + imbalance) * total / k;
  const int target = std::max(30 * k, 200);

  std::vector<Level> levels;
  Graph g = g0;
  while (g.n > target) {
    Level lv;
    Graph c = coarsen(g, lv.cmap, rng);
    if (c.n >= (int)(g.n * 0.95)) break;
    lv.g = std::move(g);
    g = std::move(c);
    levels.push_back(std::move(lv));
  }
  std::vector<int> part;
  initial_partition(g, k, max_w, part, rng);
  refine(g, k, max_w, part);
  while (!levels.empty()) {
    Level lv = std::move(levels.back());
    levels.pop_back();
    std::vector<int> fine(lv.g.n);
    for (int u = 0; u < lv.g.n; ++u) fine[u] = part[lv.cmap[u]];
    part = std::move(fine);
    refine(lv.g, k, max_w, part);
    g = std::move(lv.g);
  }
  connectivity_fixup(g, k, max_w, part);
  refine(g, k, max_w, part);
  // iterated V-cycles: re-coarsen respecting the current partition and
  // re-refine at every level; keep strict improvements
  double cut = cut_of(g, part);
  for (int vc = 0; vc < 4; ++vc) {
    auto cand = vcycle(g, k, max_w, part, target, rng);
    connectivity_fixup(g, k, max_w, cand);
    refine(g, k, max_w, cand);
    double c2 = cut_of(g, cand);
    if (c2 < cut - 1e-12) {
      part = std::move(cand);
      cut = c2;
    } else {
      break;
    }
  }
  // iterated local search: kick a random boundary blob into the
  // adjacent class, re-refine (one V-cycle), keep improvements — moves
  // FM out of local minima the gradient-like passes cannot escape
  {
    std::vector<double> wgt(k, 0.0);
    const int blob_cap = std::max(8, g.n / (16 * k));
    std::uniform_int_distribution<int> un(0, g.n - 1);
    for (int kick = 0; kick < 8; ++kick) {
      auto cand = part;
      for (int q = 0; q < k; ++q) wgt[q] = 0.0;
      for (int u = 0; u < g.n; ++u) wgt[cand[u]] += g.vwgt[u];
      int u0 = -1, tp = -1;
      for (int tries = 0; tries < 512 && u0 < 0; ++tries) {
        int x = un(rng);
        for (int e = g.xadj[x]; e < g.xadj[x + 1]; ++e)
          if (cand[g.adjncy[e]] != cand[x]) {
            u0 = x;
            tp = cand[g.adjncy[e]];
            break;
          }
      }
      if (u0 < 0) break;
      // grow a same-class BFS blob around u0 and flip it to tp
      const int p0 = cand[u0];
      std::vector<int> blob{u0}, q{u0};
      std::vector<char> inblob(g.n, 0);
      inblob[u0] = 1;
      double bw = g.vwgt[u0];
      for (size_t h = 0; h < q.size() && (int)blob.size() < blob_cap; ++h)
        for (int e = g.xadj[q[h]]; e < g.xadj[q[h] + 1]; ++e) {
          int v = g.adjncy[e];
          if (!inblob[v] && cand[v] == p0 &&
              (int)blob.size() < blob_cap &&
              wgt[tp] + bw + g.vwgt[v] <= max_w) {
            inblob[v] = 1;
            blob.push_back(v);
            q.push_back(v);
            bw += g.vwgt[v];
          }
        }
      if (wgt[tp] + bw > max_w) continue;
      for (int v : blob) cand[v] = tp;
      cand = vcycle(g, k, max_w, std::move(cand), target, rng);
      connectivity_fixup(g, k, max_w, cand);
      refine(g, k, max_w, cand);
      double c2 = cut_of(g, cand);
      if (c2 < cut - 1e-12) {
        part = std::move(cand);
        cut = c2;
      }
    }
  }
  return part;
}

}  // namespace

extern "C" {

// adjacency as CSR (xadj: n+1 ints, adjncy: m ints, adjw: m doubles or
// null for unit weights); out: n ints. Returns achieved cut weight.
double dpo_partition_multilevel(int n, const int* xadj, const int* adjncy,
                                const double* adjw, int k,
                                double imbalance, int n_restarts,
                                unsigned seed, int* out) {
  Graph g;
  g.n = n;
  g.xadj.assign(xadj, xadj + n + 1);
  g.adjncy.assign(adjncy, adjncy + xadj[n]);
  if (adjw)
    g.adjw.assign(adjw, adjw + xadj[n]);
  else
    g.adjw.assign(xadj[n], 1.0);
  g.vwgt.assign(n, 1.0);

  std::vector<int> best;
  double best_cut = 1e300;
  for (int t = 0; t < std::max(1, n_restarts); ++t) {
    auto part = multilevel_once(g, k, imbalance, seed + 977 * t);
    double cut = cut_of(g, part);
    if (cut < best_cut) {
      best_cut = cut;
      best = std::move(part);
    }
  }
  std::memcpy(out, best.data(), n * sizeof(int));
  return best_cut;
}

}  // extern "C"
