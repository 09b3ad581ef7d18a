// Host-side native greedy solver ("limited mode") over SoA candidate arrays.
//
// Semantics mirror the Python golden (inferno_amd/solver/greedy.py), which
// mirrors the reference's pkg/solver/greedy.go:35-341: per-server sorted
// candidate lists with delta-regret ordering, iterative allocation with
// leftmost re-insertion, priority groups, and the three best-effort
// saturation policies. This is the hot-path backend the GPU sweep feeds
// (Python-object materialization dominated limited
// mode at fleet scale); the Python implementation remains the differential
// oracle.
//
// Inputs are flat arrays (one row per FEASIBLE candidate, grouped by server
// in ascending server order, candidates within a server pre-sorted by value
// ascending with ties in accelerator-name order — exactly the order
// greedy.py sees):
//   cand_value[f32]   ordering value (cost + transition penalty)
//   cand_acc_type[i32] accelerator-type index (-1 = zero-load empty alloc:
//                      the entry is dropped like acc-lookup-miss in Python)
//   cand_units[i32]   units per replica (numInstances * multiplicity)
//   cand_replicas[i32] desired replicas
//   seg_start[i32]    [n_servers+1] candidate segment per server
//   srv_priority[i32] per server
//   capacity[i32]     [n_types] available units per accelerator type
//                      (mutated in place)
// Outputs (per server):
//   out_cand[i32]     winning candidate row (-1 = unallocated)
//   out_replicas[i32] granted replicas (may be < cand_replicas under
//                      best-effort partial allocation)
#include <algorithm>
#include <cstdint>
#include <vector>

namespace {

constexpr double kMaxF32 = 3.4028234663852886e38;

struct Entry {
  int srv;
  int cur;       // index into the server's candidate segment
  double delta;  // regret to the next-best candidate
  int seq;       // tie-break: original (server-name) order
};

struct Ctx {
  const float *value;
  const int *acc_type;
  const int *units;
  const int *replicas;
  const int *seg_start;
  const int *priority;
  int *capacity;
  int n_types;
  int *out_cand;
  int *out_replicas;

  int seg_len(int srv) const { return seg_start[srv + 1] - seg_start[srv]; }
  double val(int srv, int idx) const {
    return (double)value[seg_start[srv] + idx];
  }
};

// ordering key (priority asc, delta desc, value desc), stable on seq —
// matches greedy.py _order_key + Python stable sort / bisect_left insertion
struct Key {
  int priority;
  double neg_delta;
  double neg_value;
};

inline Key key_of(const Ctx &c, const Entry &e) {
  return Key{c.priority[e.srv], -e.delta, -c.val(e.srv, e.cur)};
}

inline bool key_less(const Key &a, const Key &b) {
  if (a.priority != b.priority) return a.priority < b.priority;
  if (a.neg_delta != b.neg_delta) return a.neg_delta < b.neg_delta;
  return a.neg_value < b.neg_value;
}

// greedy.py _allocate (greedy.go:107-166). Entries arrive ordered; returns
// the unallocated entries in the order they fell out.
std::vector<Entry> allocate(Ctx &c, std::vector<Entry> entries) {
  std::vector<Key> keys;
  keys.reserve(entries.size());
  for (const auto &e : entries) keys.push_back(key_of(c, e));
  std::vector<Entry> unallocated;
  size_t head = 0;  // pop-front without memmove
  while (head < entries.size()) {
    Entry top = entries[head];
    Key _k = keys[head];
    (void)_k;
    ++head;
    const int len = c.seg_len(top.srv);
    if (len == 0) continue;
    const int cand = c.seg_start[top.srv] + top.cur;
    const int t = c.acc_type[cand];
    if (t < 0) continue;  // zero-load "" accelerator: dropped (greedy.py:108)
    const long long count =
        (long long)c.replicas[cand] * (long long)c.units[cand];
    if ((long long)c.capacity[t] >= count) {
      c.capacity[t] -= (int)count;
      c.out_cand[top.srv] = cand;
      c.out_replicas[top.srv] = c.replicas[cand];
    } else {
      top.cur += 1;
      if (top.cur + 1 < len) {
        top.delta = c.val(top.srv, top.cur + 1) - c.val(top.srv, top.cur);
      } else if (top.cur == len) {
        unallocated.push_back(top);
        continue;
      } else {
        top.delta = kMaxF32;
      }
      // leftmost insertion among equal keys within the remaining tail
      Key k = key_of(c, top);
      auto pos = std::lower_bound(keys.begin() + head, keys.end(), k, key_less);
      size_t i = (size_t)(pos - keys.begin());
      keys.insert(keys.begin() + i, k);
      entries.insert(entries.begin() + i, top);
    }
  }
  return unallocated;
}

// greedy.py allocate_maximally (greedy.go:194-223)
void allocate_maximally(Ctx &c, const std::vector<Entry> &entries) {
  for (const auto &e : entries) {
    const int len = c.seg_len(e.srv);
    for (int i = 0; i < len; ++i) {
      const int cand = c.seg_start[e.srv] + i;
      const int t = c.acc_type[cand];
      if (t < 0) continue;
      const int u = c.units[cand];
      if (u <= 0) continue;
      int max_rep = c.capacity[t] / u;
      if (max_rep > c.replicas[cand]) max_rep = c.replicas[cand];
      if (max_rep > 0) {
        c.out_cand[e.srv] = cand;
        c.out_replicas[e.srv] = max_rep;
        c.capacity[t] -= max_rep * u;
        break;
      }
    }
  }
}

// greedy.py allocate_equally (greedy.go:239-316): round-robin tickets
void allocate_equally(Ctx &c, const std::vector<Entry> &entries) {
  struct Ticket {
    bool live = true;
    bool active = false;
    int acc_type = -1;
    int units = 0;
    int granted = 0;
    int cand = -1;
  };
  const size_t n = entries.size();
  std::vector<Ticket> tickets(n);
  size_t live = n;
  while (live > 0) {
    for (size_t j = 0; j < n; ++j) {
      Ticket &tk = tickets[j];
      if (!tk.live) continue;
      const Entry &e = entries[j];
      if (!tk.active) {
        const int len = c.seg_len(e.srv);
        for (int i = 0; i < len; ++i) {
          const int cand = c.seg_start[e.srv] + i;
          const int t = c.acc_type[cand];
          if (t < 0) continue;
          const int u = c.units[cand];
          if (u > 0 && c.capacity[t] >= u) {
            tk.active = true;
            tk.acc_type = t;
            tk.units = u;
            tk.cand = cand;
            break;
          }
        }
        if (!tk.active) {
          tk.live = false;
          --live;
          continue;
        }
      }
      const int avail_rep = c.capacity[tk.acc_type] / tk.units;
      const int want = c.replicas[tk.cand];
      if ((avail_rep < want ? avail_rep : want) > 0) {
        tk.granted += 1;
        c.capacity[tk.acc_type] -= tk.units;
      } else {
        tk.live = false;
        --live;
      }
    }
  }
  for (size_t j = 0; j < n; ++j) {
    const Ticket &tk = tickets[j];
    if (tk.granted > 0) {
      c.out_cand[entries[j].srv] = tk.cand;
      c.out_replicas[entries[j].srv] = tk.granted;
    }
  }
}

void best_effort(Ctx &c, const std::vector<Entry> &unallocated, int policy) {
  // policy: 0 None, 1 PriorityExhaustive, 2 PriorityRoundRobin, 3 RoundRobin
  if (policy == 1) {
    allocate_maximally(c, unallocated);
  } else if (policy == 2) {
    size_t i = 0;
    while (i < unallocated.size()) {
      size_t j = i + 1;
      const int prio = c.priority[unallocated[i].srv];
      while (j < unallocated.size() && c.priority[unallocated[j].srv] == prio) ++j;
      allocate_equally(
          c, std::vector<Entry>(unallocated.begin() + i, unallocated.begin() + j));
      i = j;
    }
  } else if (policy == 3) {
    allocate_equally(c, unallocated);
  }
}

}  // namespace

extern "C" int wva_greedy_solve(
    int n_servers, int n_types, const float *cand_value, const int *cand_acc_type,
    const int *cand_units, const int *cand_replicas, const int *seg_start,
    const int *srv_priority, int *capacity, int delayed_best_effort, int policy,
    int *out_cand, int *out_replicas) {
  Ctx c{cand_value, cand_acc_type, cand_units,  cand_replicas, seg_start,
        srv_priority, capacity,     n_types,     out_cand,      out_replicas};
  for (int s = 0; s < n_servers; ++s) {
    out_cand[s] = -1;
    out_replicas[s] = 0;
  }
  // build + order entries (greedy.py solve_greedy:49-70)
  std::vector<Entry> entries;
  entries.reserve(n_servers);
  for (int s = 0; s < n_servers; ++s) {
    const int len = c.seg_len(s);
    if (len == 0) continue;
    Entry e{s, 0, 0.0, (int)entries.size()};
    e.delta = (len > 1) ? (c.val(s, 1) - c.val(s, 0)) : kMaxF32;
    entries.push_back(e);
  }
  std::stable_sort(entries.begin(), entries.end(), [&](const Entry &a, const Entry &b) {
    return key_less(key_of(c, a), key_of(c, b));
  });

  if (delayed_best_effort) {
    auto unalloc = allocate(c, std::move(entries));
    best_effort(c, unalloc, policy);
  } else {
    size_t i = 0;
    while (i < entries.size()) {
      size_t j = i + 1;
      const int prio = c.priority[entries[i].srv];
      while (j < entries.size() && c.priority[entries[j].srv] == prio) ++j;
      auto unalloc = allocate(
          c, std::vector<Entry>(entries.begin() + i, entries.begin() + j));
      best_effort(c, unalloc, policy);
      i = j;
    }
  }
  return 0;
}
