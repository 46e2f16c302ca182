// Native event scheduler for the batched gossip engine (CPU, no GPU deps).
//
// Bit-exact C++ replica of gossipy_amd/engine/schedule.py — same splitmix64
// tape (engine/rng.py), same draw order, same slot free-list discipline —
// but ~100x faster, and it emits the round as FLAT packed arrays (one
// upload per round) for the GPU round executor instead of per-tick Python
// objects. Python<->C++ equivalence is enforced by
// tests/test_native_sched.py on every CPU test run.
//
// Output layout per round (all int32 numpy arrays; `delta` ticks):
//   snap_nodes/snap_slots + snap_tptr[delta+1]      sub-phase A per tick
//   recv_nodes + recv_nptr (global CSR over msgs)
//     + recv_tptr[delta+1] (tick -> receiver range)  sub-phase B
//   del_slots/del_owners/reply_slots  (aligned with recv_nptr[-1] msgs)
//   pull_nodes/pull_slots + pull_tptr[delta+1]       PULL snapshots
//   rep_* (same structure as recv_*)                 sub-phase C
//   eval_nodes; scalars sent/failed/total_size/n_slots(high water)

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <map>
#include <deque>
#include <functional>
#include <unordered_map>
#include <chrono>
#include <cstdio>
#include <unordered_set>
#include <vector>

namespace py = pybind11;

// ---------------------------------------------------------------------------
// tape (must match engine/rng.py exactly)
// ---------------------------------------------------------------------------

static inline uint64_t splitmix64(uint64_t x)
{
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

struct Stream {
    uint64_t key;
    uint64_t off = 0;
    explicit Stream(uint64_t k) : key(k) {}
    inline uint64_t raw() { return splitmix64(key + off++); }
    inline double rnd()
    {
        return (double)(raw() >> 11) * (1.0 / 9007199254740992.0);
    }
    inline int64_t integers(int64_t lo, int64_t hi)
    {
        return lo + (int64_t)std::floor(rnd() * (double)(hi - lo));
    }
    inline double normal(double mu, double sigma)
    {
        double u1 = std::max(rnd(), 1e-300);
        double u2 = rnd();
        return mu + sigma * std::sqrt(-2.0 * std::log(u1)) *
                        std::cos(2.0 * M_PI * u2);
    }
};

enum Purpose {
    TIMEOUT = 0, PEER = 1, DROP = 2, ONLINE = 3, DELAY = 4,
    EVAL = 5, INIT = 6, DATA = 7, TOKEN = 8, MISC = 9, PART = 10,
    SAMPLE = 11
};

static inline uint64_t stream_key(uint64_t seed, uint64_t purpose, uint64_t t,
                                  uint64_t extra)
{
    uint64_t k = seed;
    k = splitmix64(k ^ purpose);
    k = splitmix64(k ^ t);
    k = splitmix64(k ^ extra);
    return k;
}

// ---------------------------------------------------------------------------
// config / protocol enums (numbering matches gossipy_amd.core)
// ---------------------------------------------------------------------------

enum Proto { PUSH = 1, PULL = 2, PUSH_PULL = 3 };
enum DelayKind { D_CONST = 0, D_UNIFORM = 1, D_LINEAR = 2 };

struct Msg {  // matches the python pending tuple
    int32_t receiver;
    int32_t slot;       // -1 for PULL requests
    int32_t reply_flag; // -2 = wants reply, -1 = plain
    bool is_pull;
    int32_t sender;
    int32_t pid;        // partition id (-1 when n_parts == 0)
};


// ---------------------------------------------------------------------------
// Launch-group merging (mirrors BatchedGossipSimulator._merge_flat_groups):
// fuse conflict-free adjacent ticks into single launch groups. Returns the
// tick-index boundaries [0, ..., delta]. Epoch-stamped membership arrays give
// O(total events) with no per-group clearing.
// ---------------------------------------------------------------------------
static std::vector<int32_t> compute_merge_bounds(
    int64_t n_nodes, int64_t n_slots,
    const std::vector<int32_t>& snap_nodes, const std::vector<int32_t>& snap_slots,
    const std::vector<int32_t>& snap_tptr,
    const std::vector<int32_t>& recv_nodes, const std::vector<int32_t>& recv_nptr,
    const std::vector<int32_t>& recv_tptr,
    const std::vector<int32_t>& del_slots, const std::vector<int32_t>& reply_slots,
    const std::vector<int32_t>& pull_nodes, const std::vector<int32_t>& pull_slots,
    const std::vector<int32_t>& pull_tptr,
    const std::vector<int32_t>& rep_nodes, const std::vector<int32_t>& rep_nptr,
    const std::vector<int32_t>& rep_tptr, const std::vector<int32_t>& rep_slots)
{
    int64_t delta = (int64_t)snap_tptr.size() - 1;
    std::vector<int32_t> bounds{0};
    if (delta <= 1) {
        if (delta == 1) bounds.push_back(1);
        return bounds;
    }
    std::vector<int32_t> node_mut(n_nodes, -1), node_pull(n_nodes, -1);
    std::vector<int32_t> slot_read(n_slots, -1), slot_write(n_slots, -1),
        slot_late(n_slots, -1);
    int32_t gid = 0;
    auto tick_ok = [&](int64_t t) -> bool {
        // candidate tick's reads/mutations vs the group's accumulated sets
        for (int32_t i = snap_tptr[t]; i < snap_tptr[t + 1]; ++i) {
            if (node_mut[snap_nodes[i]] == gid) return false;        // touched & g_mut
            if (slot_read[snap_slots[i]] == gid) return false;       // writes & g_reads
            if (slot_write[snap_slots[i]] == gid) return false;      // writes & g_writes
        }
        for (int32_t i = pull_tptr[t]; i < pull_tptr[t + 1]; ++i) {
            if (node_mut[pull_nodes[i]] == gid) return false;
            if (slot_read[pull_slots[i]] == gid) return false;
            if (slot_write[pull_slots[i]] == gid) return false;
        }
        for (int32_t r = recv_tptr[t]; r < recv_tptr[t + 1]; ++r) {
            int32_t x = recv_nodes[r];
            if (node_mut[x] == gid) return false;                    // touched & g_mut
            if (node_pull[x] == gid) return false;                   // recv & g_pulls
            for (int32_t d = recv_nptr[r]; d < recv_nptr[r + 1]; ++d) {
                if (slot_late[del_slots[d]] == gid) return false;    // dreads & g_late
                int32_t rw = reply_slots[d];
                if (rw >= 0 && (slot_read[rw] == gid || slot_write[rw] == gid))
                    return false;                                    // writes & ...
            }
        }
        for (int32_t r = rep_tptr[t]; r < rep_tptr[t + 1]; ++r) {
            if (node_mut[rep_nodes[r]] == gid) return false;
        }
        return true;
    };
    auto tick_add = [&](int64_t t) {
        for (int32_t i = snap_tptr[t]; i < snap_tptr[t + 1]; ++i)
            slot_write[snap_slots[i]] = gid;
        for (int32_t i = pull_tptr[t]; i < pull_tptr[t + 1]; ++i) {
            node_pull[pull_nodes[i]] = gid;
            slot_write[pull_slots[i]] = gid;
            slot_late[pull_slots[i]] = gid;
        }
        for (int32_t r = recv_tptr[t]; r < recv_tptr[t + 1]; ++r) {
            node_mut[recv_nodes[r]] = gid;
            for (int32_t d = recv_nptr[r]; d < recv_nptr[r + 1]; ++d) {
                slot_read[del_slots[d]] = gid;
                int32_t rw = reply_slots[d];
                if (rw >= 0) { slot_write[rw] = gid; slot_late[rw] = gid; }
            }
        }
        for (int32_t r = rep_tptr[t]; r < rep_tptr[t + 1]; ++r) {
            node_mut[rep_nodes[r]] = gid;
            for (int32_t d = rep_nptr[r]; d < rep_nptr[r + 1]; ++d)
                slot_read[rep_slots[d]] = gid;
        }
    };
    tick_add(0);
    for (int64_t t = 1; t < delta; ++t) {
        if (!tick_ok(t)) {
            bounds.push_back((int32_t)t);
            ++gid;
        }
        tick_add(t);
    }
    bounds.push_back((int32_t)delta);
    return bounds;
}


// ---------------------------------------------------------------------------
// Entry-level launch packing with per-node deferral (bit-exact port of
// BatchedGossipSimulator._pack_flat — parity enforced by
// tests/test_native_sched.py). Deliveries to one receiver coalesce across
// ticks into one CSR row; snapshots of receiving nodes embed as per-delivery
// reply writes; same-group produce->consume pairs split across two deliver
// launches; events needing a third level DEFER their node and replay (in
// global schedule order — causality) after the group closes, so burst
// chains no longer fragment the round. Returns packed arrays in the
// executors' format.
// ---------------------------------------------------------------------------
static py::dict pack_round(
    const std::vector<int32_t>& snap_nodes, const std::vector<int32_t>& snap_slots,
    const std::vector<int32_t>& snap_tptr,
    const std::vector<int32_t>& recv_nodes, const std::vector<int32_t>& recv_nptr,
    const std::vector<int32_t>& recv_tptr,
    const std::vector<int32_t>& del_slots, const std::vector<int32_t>& reply_slots,
    const std::vector<int32_t>& del_pids,
    const std::vector<int32_t>& pull_nodes, const std::vector<int32_t>& pull_slots,
    const std::vector<int32_t>& pull_tptr,
    const std::vector<int32_t>& rep_nodes, const std::vector<int32_t>& rep_nptr,
    const std::vector<int32_t>& rep_tptr, const std::vector<int32_t>& rep_slots,
    const std::vector<int32_t>& rep_pids,
    const std::vector<int32_t>& del_owners,
    const std::vector<int32_t>& rep_owners,
    int64_t n_nodes, int64_t n_slots)
{
    int64_t delta = (int64_t)snap_tptr.size() - 1;
    bool has_pid = !del_pids.empty() || !rep_pids.empty();
    bool has_own = !del_owners.empty() || !rep_owners.empty();

    struct Ev { int32_t slot, reply, pid, own, next; };
    struct Row { int32_t node, head, tail; };
    struct Out {
        std::vector<int32_t> snap_nodes, snap_slots, snap_tptr{0};
        std::vector<int32_t> recv_nodes, recv_nptr{0}, recv_tptr{0};
        std::vector<int32_t> del_slots, reply_slots, del_pids, del_owners;
        std::vector<int32_t> rep_nodes, rep_nptr{0}, rep_tptr{0};
        std::vector<int32_t> rep_slots, rep_reply_slots, rep_pids, rep_owners;
        std::vector<int32_t> pull_tptr{};
    } o;
    o.snap_nodes.reserve(snap_nodes.size() + pull_nodes.size());
    o.snap_slots.reserve(snap_slots.size() + pull_slots.size());
    o.recv_nodes.reserve(recv_nodes.size());
    o.del_slots.reserve(del_slots.size() + rep_slots.size());
    o.reply_slots.reserve(del_slots.size() + rep_slots.size());

    // group-local state in epoch-stamped flat arrays (stamp = group id)
    int32_t n_slots_cap = (int32_t)n_slots;
    int32_t n_nodes_cap = (int32_t)n_nodes;
    std::vector<int32_t> t_stamp(n_slots_cap, -1);           // touched
    std::vector<int32_t> w_stamp(n_slots_cap, -1);           // written
    std::vector<int8_t> w_lvl(n_slots_cap, 0);               // 1/2/3, 9=pending
    std::vector<int32_t> l_stamp(n_nodes_cap, -1);           // latest row
    std::vector<int8_t> l_kind(n_nodes_cap, 0);
    std::vector<int32_t> l_idx(n_nodes_cap, 0);
    std::vector<int32_t> r2_stamp(n_nodes_cap, -1), r2_idx(n_nodes_cap, 0);
    std::vector<int32_t> r3_stamp(n_nodes_cap, -1), r3_idx(n_nodes_cap, 0);
    int32_t gid = 0;

    std::vector<std::pair<int32_t, int32_t>> g_snap;
    std::vector<Ev> pool;
    pool.reserve(del_slots.size() + rep_slots.size() + 16);
    std::vector<Row> l2, l3;

    // deferral state: node flags + (node, event) queue in schedule order
    struct DEv { int32_t kind, slot, reply, pid, own; };  // kind 0=snap 1=deliv
    std::vector<int8_t> is_def(n_nodes_cap, 0);
    std::vector<std::pair<int32_t, DEv>> deferred;

    struct Packer {
        std::vector<std::pair<int32_t, int32_t>>& g_snap;
        std::vector<Ev>& pool;
        std::vector<Row>& l2;
        std::vector<Row>& l3;
        std::vector<int32_t>& t_stamp;
        std::vector<int32_t>& w_stamp;
        std::vector<int8_t>& w_lvl;
        std::vector<int32_t>& l_stamp;
        std::vector<int8_t>& l_kind;
        std::vector<int32_t>& l_idx;
        std::vector<int32_t>& r2_stamp;
        std::vector<int32_t>& r2_idx;
        std::vector<int32_t>& r3_stamp;
        std::vector<int32_t>& r3_idx;
        int32_t& gid;
        std::vector<int8_t>& is_def;
        std::vector<std::pair<int32_t, DEv>>& deferred;
        bool in_replay = false;
        Out& o;
        bool has_pid, has_own;

        void append_ev(Row& row, int32_t slot, int32_t reply, int32_t pid,
                       int32_t own)
        {
            int32_t idx = (int32_t)pool.size();
            pool.push_back({slot, reply, pid, own, -1});
            if (row.tail >= 0) pool[row.tail].next = idx;
            else row.head = idx;
            row.tail = idx;
        }

        void flush()
        {
            for (auto& p : g_snap) {
                o.snap_nodes.push_back(p.first);
                o.snap_slots.push_back(p.second);
            }
            o.snap_tptr.push_back((int32_t)o.snap_nodes.size());
            for (auto& row : l2) {
                o.recv_nodes.push_back(row.node);
                for (int32_t e = row.head; e >= 0; e = pool[e].next) {
                    o.del_slots.push_back(pool[e].slot);
                    o.reply_slots.push_back(pool[e].reply);
                    if (has_pid) o.del_pids.push_back(pool[e].pid);
                    if (has_own) o.del_owners.push_back(pool[e].own);
                }
                o.recv_nptr.push_back((int32_t)o.del_slots.size());
            }
            o.recv_tptr.push_back((int32_t)o.recv_nodes.size());
            for (auto& row : l3) {
                o.rep_nodes.push_back(row.node);
                for (int32_t e = row.head; e >= 0; e = pool[e].next) {
                    o.rep_slots.push_back(pool[e].slot);
                    o.rep_reply_slots.push_back(pool[e].reply);
                    if (has_pid) o.rep_pids.push_back(pool[e].pid);
                    if (has_own) o.rep_owners.push_back(pool[e].own);
                }
                o.rep_nptr.push_back((int32_t)o.rep_slots.size());
            }
            o.rep_tptr.push_back((int32_t)o.rep_nodes.size());
            g_snap.clear(); l2.clear(); l3.clear();
            ++gid;
        }

        void defer(int32_t node, const DEv& ev)
        {
            is_def[node] = 1;
            deferred.emplace_back(node, ev);
            if (ev.kind == 0) {
                w_stamp[ev.slot] = gid; w_lvl[ev.slot] = 9;
                t_stamp[ev.slot] = gid;
            } else {
                t_stamp[ev.slot] = gid;
                if (ev.reply >= 0) {
                    w_stamp[ev.reply] = gid; w_lvl[ev.reply] = 9;
                    t_stamp[ev.reply] = gid;
                }
            }
        }

        void close()
        {
            if (in_replay) {
                flush();
                return;
            }
            flush();
            while (!deferred.empty()) {
                in_replay = true;
                std::vector<std::pair<int32_t, DEv>> dl;
                dl.swap(deferred);
                for (auto& p : dl) is_def[p.first] = 0;
                for (auto& p : dl) {
                    const DEv& ev = p.second;
                    if (ev.kind == 0) place_snap(p.first, ev.slot);
                    else
                        place_delivery(p.first, ev.slot, ev.reply, ev.pid,
                                       ev.own);
                }
                in_replay = false;
                if (!deferred.empty()) flush();  // next pass = fresh group
            }
        }

        void place_snap(int32_t node, int32_t slot)
        {
            if (is_def[node]) { defer(node, {0, slot, -1, -1, -1}); return; }
            if (t_stamp[slot] == gid) {
                close();
                if (is_def[node]) {  // replay re-deferred this node
                    defer(node, {0, slot, -1, -1, -1});
                    return;
                }
            }
            if (l_stamp[node] != gid) {
                g_snap.emplace_back(node, slot);
                w_stamp[slot] = gid; w_lvl[slot] = 1;
            } else {
                auto& lst = (l_kind[node] == 2) ? l2 : l3;
                Ev& last = pool[lst[l_idx[node]].tail];
                if (last.reply >= 0) {
                    defer(node, {0, slot, -1, -1, -1});
                    return;
                }
                last.reply = slot;
                w_stamp[slot] = gid;
                w_lvl[slot] = (l_kind[node] == 2) ? 2 : 3;
            }
            t_stamp[slot] = gid;
        }

        void place_delivery(int32_t node, int32_t slot, int32_t reply,
                            int32_t pid, int32_t own)
        {
            if (is_def[node]) {
                defer(node, {1, slot, reply, pid, own});
                return;
            }
            int lvl = (w_stamp[slot] == gid) ? w_lvl[slot] : 0;
            if (lvl >= 3 || (reply >= 0 && t_stamp[reply] == gid)) {
                defer(node, {1, slot, reply, pid, own});
                return;
            }
            bool in_l3 = (l_stamp[node] == gid && l_kind[node] == 3);
            int wl;
            if (in_l3) {
                append_ev(l3[l_idx[node]], slot, reply, pid, own);
                wl = 3;
            } else if (lvl >= 2) {
                int32_t ri;
                if (r3_stamp[node] != gid) {
                    ri = (int32_t)l3.size();
                    r3_stamp[node] = gid; r3_idx[node] = ri;
                    l3.push_back({node, -1, -1});
                } else {
                    ri = r3_idx[node];
                }
                append_ev(l3[ri], slot, reply, pid, own);
                l_stamp[node] = gid; l_kind[node] = 3; l_idx[node] = ri;
                wl = 3;
            } else {
                int32_t ri;
                if (r2_stamp[node] != gid) {
                    ri = (int32_t)l2.size();
                    r2_stamp[node] = gid; r2_idx[node] = ri;
                    l2.push_back({node, -1, -1});
                    l_stamp[node] = gid; l_kind[node] = 2; l_idx[node] = ri;
                } else {
                    ri = r2_idx[node];
                }
                append_ev(l2[ri], slot, reply, pid, own);
                wl = 2;
            }
            t_stamp[slot] = gid;
            if (reply >= 0) {
                w_stamp[reply] = gid; w_lvl[reply] = (int8_t)wl;
                t_stamp[reply] = gid;
            }
        }
    } P{g_snap, pool, l2, l3, t_stamp, w_stamp, w_lvl, l_stamp, l_kind,
       l_idx, r2_stamp, r2_idx, r3_stamp, r3_idx, gid, is_def, deferred,
       false, o, has_pid, has_own};

    for (int64_t t = 0; t < delta; ++t) {
        for (int32_t i = snap_tptr[t]; i < snap_tptr[t + 1]; ++i)
            P.place_snap(snap_nodes[i], snap_slots[i]);
        for (int32_t r = recv_tptr[t]; r < recv_tptr[t + 1]; ++r) {
            int32_t x = recv_nodes[r];
            for (int32_t d = recv_nptr[r]; d < recv_nptr[r + 1]; ++d) {
                int32_t pid =
                    (has_pid && !del_pids.empty()) ? del_pids[d] : -1;
                int32_t own =
                    (has_own && !del_owners.empty()) ? del_owners[d] : -1;
                P.place_delivery(x, del_slots[d], reply_slots[d], pid, own);
            }
        }
        for (int32_t i = pull_tptr[t]; i < pull_tptr[t + 1]; ++i)
            P.place_snap(pull_nodes[i], pull_slots[i]);
        for (int32_t r = rep_tptr[t]; r < rep_tptr[t + 1]; ++r) {
            int32_t x = rep_nodes[r];
            for (int32_t d = rep_nptr[r]; d < rep_nptr[r + 1]; ++d) {
                int32_t pid =
                    (has_pid && !rep_pids.empty()) ? rep_pids[d] : -1;
                int32_t own =
                    (has_own && !rep_owners.empty()) ? rep_owners[d] : -1;
                P.place_delivery(x, rep_slots[d], -1, pid, own);
            }
        }
    }
    int guard = 0;
    while ((!g_snap.empty() || !l2.empty() || !l3.empty() ||
            !deferred.empty()) && guard < 100000) {
        P.close();
        ++guard;
    }

    int32_t n_groups = (int32_t)o.snap_tptr.size() - 1;
    o.pull_tptr.assign((size_t)n_groups + 1, 0);
    py::dict pd;
    auto arr = [](std::vector<int32_t>& v) {
        auto a = py::array_t<int32_t>((py::ssize_t)v.size());
        std::copy(v.begin(), v.end(), a.mutable_data());
        return a;
    };
    std::vector<int32_t> empty;
    pd["snap_nodes"] = arr(o.snap_nodes);
    pd["snap_slots"] = arr(o.snap_slots);
    pd["snap_tptr"] = arr(o.snap_tptr);
    pd["recv_nodes"] = arr(o.recv_nodes);
    pd["recv_nptr"] = arr(o.recv_nptr);
    pd["recv_tptr"] = arr(o.recv_tptr);
    pd["del_slots"] = arr(o.del_slots);
    pd["reply_slots"] = arr(o.reply_slots);
    pd["del_pids"] = arr(o.del_pids);
    pd["del_owners"] = arr(o.del_owners);
    pd["pull_nodes"] = arr(empty);
    pd["pull_slots"] = arr(empty);
    pd["pull_tptr"] = arr(o.pull_tptr);
    pd["rep_nodes"] = arr(o.rep_nodes);
    pd["rep_nptr"] = arr(o.rep_nptr);
    pd["rep_tptr"] = arr(o.rep_tptr);
    pd["rep_slots"] = arr(o.rep_slots);
    pd["rep_reply_slots"] = arr(o.rep_reply_slots);
    pd["rep_pids"] = arr(o.rep_pids);
    pd["rep_owners"] = arr(o.rep_owners);
    return pd;
}

class NativeScheduler {
public:
    NativeScheduler(int64_t n_nodes, int64_t delta, int proto,
                    int64_t model_size, double drop_prob, double online_prob,
                    int delay_kind, int64_t dmin, int64_t dmax,
                    double timexunit, int64_t overhead, bool sync,
                    double sampling_eval, uint64_t seed,
                    py::object peers_indptr, py::object peers_indices,
                    int64_t n_parts = 0, bool sampled = false)
        : n_(n_nodes), delta_(delta), proto_(proto), model_size_(model_size),
          drop_(drop_prob), online_(online_prob), dkind_(delay_kind),
          dmin_(dmin), dmax_(dmax), timexunit_(timexunit), overhead_(overhead),
          sync_(sync), sampling_eval_(sampling_eval), seed_(seed),
          n_parts_(n_parts), sampled_(sampled)
    {
        pend_init();
        Stream g(stream_key(seed_, TIMEOUT, 0, 0));
        deltas_.resize(n_);
        if (sync_) {
            for (int64_t i = 0; i < n_; ++i)
                deltas_[i] = g.integers(0, delta_);
        } else {
            for (int64_t i = 0; i < n_; ++i)
                deltas_[i] = std::max<int64_t>(
                    1, (int64_t)g.normal((double)delta_, (double)delta_ / 10.0));
        }
        if (!peers_indptr.is_none()) {
            auto ip = peers_indptr.cast<py::array_t<int64_t>>();
            auto ix = peers_indices.cast<py::array_t<int64_t>>();
            indptr_.assign(ip.data(), ip.data() + ip.size());
            indices_.assign(ix.data(), ix.data() + ix.size());
        }
        slot_owner_.resize(64);
        if (sync_) {  // static offsets: bucket nodes by firing phase once
            fire_buckets_.resize(delta_);
            for (int64_t i = 0; i < n_; ++i)
                fire_buckets_[deltas_[i]].push_back((int32_t)i);
        }
    }

    py::dict next_round(int64_t r);

protected:
    int64_t n_, delta_;
    int proto_;
    int64_t model_size_;
    double drop_, online_;
    int dkind_;
    int64_t dmin_, dmax_;
    double timexunit_;
    int64_t overhead_;
    bool sync_;
    double sampling_eval_;
    uint64_t seed_;
    int64_t n_parts_ = 0;
    bool sampled_ = false;
    std::vector<int64_t> deltas_;
    std::vector<std::vector<int32_t>> fire_buckets_;
    std::vector<int64_t> indptr_, indices_;
    std::unordered_map<int64_t, std::vector<Msg>> pending_;
    // delay-bounded ring over in-flight messages: slot = tick & mask.
    // Enabled when the maximum possible delay is known and small (const /
    // uniform / linear with the config's model size), which removes the
    // per-message hash — the dominant host cost at 10k+ nodes. Exact-order
    // preserving: a slot holds one tick's messages in insertion order, and
    // take() swaps them out so same-tick re-insertions (replies) land in a
    // fresh vector just as the map's erase/reinsert did.
    size_t ring_mask_ = 0;
    std::vector<int64_t> ring_tick_;
    std::vector<std::vector<Msg>> ring_;
    // per-tick receiver-row scratch, epoch-stamped by (2t | phase) so the
    // 100-tick loop never allocates a map or clears an n-sized array
    // (the per-tick unordered_map + per-row vectors were ~half the loop
    // cost at 50k nodes)
    std::vector<int64_t> row_stamp_;
    std::vector<int32_t> row_idx_;
public:
    // lean emission: the single-rank fast path consumes ONLY the packed
    // schedule (+ tick pointers for accounting), so the full per-event
    // arrays and the multi-rank merge bounds are skipped — at 50k nodes
    // that halves the emit phase
    void set_lean(bool v) { lean_ = v; }
protected:
    bool lean_ = false;
    struct DEv { int32_t v0, v1, v2, v3, next; };
    std::vector<DEv> ev_pool_;
    std::vector<int32_t> row_head_, row_tail_;

    void pend_init()
    {
        int64_t bound;
        switch (dkind_) {
        case D_CONST: bound = dmin_; break;
        case D_UNIFORM: bound = dmax_; break;
        default:
            bound = (int64_t)(timexunit_ * (double)model_size_) + overhead_;
        }
        if (bound >= 0 && bound < 65536) {
            size_t cap = 1;
            while (cap < (size_t)bound + 2) cap <<= 1;
            ring_mask_ = cap - 1;
            ring_tick_.assign(cap, INT64_MIN);
            ring_.resize(cap);
        }
        row_stamp_.assign((size_t)n_, INT64_MIN);
        row_idx_.assign((size_t)n_, 0);
        slot_reuse_lag_ = std::max<int64_t>(32, delta_);
    }

    void pend_push(int64_t t, const Msg& m)
    {
        if (ring_mask_) {
            size_t sl = (size_t)t & ring_mask_;
            if (ring_tick_[sl] != t) {
                ring_[sl].clear();
                ring_tick_[sl] = t;
            }
            ring_[sl].push_back(m);
        } else {
            pending_[t].push_back(m);
        }
    }

    std::vector<Msg> pend_take(int64_t t)
    {
        std::vector<Msg> out;
        if (ring_mask_) {
            size_t sl = (size_t)t & ring_mask_;
            if (ring_tick_[sl] == t) {
                out.swap(ring_[sl]);
                ring_tick_[sl] = INT64_MIN;
            }
        } else {
            auto it = pending_.find(t);
            if (it != pending_.end()) {
                out = std::move(it->second);
                pending_.erase(it);
            }
        }
        return out;
    }
    // consumed slots re-enter circulation only slot_reuse_lag_ ticks after
    // the tick that freed them (FIFO), mirroring Scheduler.SLOT_REUSE_LAG.
    // The effective lag is max(32, delta): launch groups never span a
    // round, so a lag of >= one full round makes every packed group
    // structurally alias-free (the packers' touched-set hazard checks
    // remain as defense-in-depth). Set in pend_init() once delta_ is known.
    int64_t slot_reuse_lag_ = 32;
    std::deque<std::pair<int64_t, int32_t>> reuse_q_;
    int64_t next_slot_ = 0;
    std::vector<int32_t> slot_owner_;

    int32_t alloc_slot(int32_t owner, int64_t t)
    {
        int32_t s;
        if (!reuse_q_.empty() && reuse_q_.front().first + slot_reuse_lag_ <= t) {
            s = reuse_q_.front().second;
            reuse_q_.pop_front();
        } else {
            s = (int32_t)next_slot_++;
            if (next_slot_ > (int64_t)slot_owner_.size())
                slot_owner_.resize(slot_owner_.size() * 2);
        }
        slot_owner_[s] = owner;
        return s;
    }

    int64_t delay_for(Stream& g, int64_t size)
    {
        switch (dkind_) {
        case D_CONST: return dmin_;
        case D_UNIFORM: return g.integers(dmin_, dmax_ + 1);
        default: return (int64_t)(timexunit_ * (double)size) + overhead_;
        }
    }

    // fresh partition id for a reply (keyed on (t, replier) like the
    // python scheduler's _reply_pid)
    int32_t reply_pid(int64_t t, int32_t replier)
    {
        if (n_parts_ > 0) {
            Stream g(stream_key(seed_, PART, (uint64_t)t,
                                (uint64_t)(1 + replier)));
            return (int32_t)g.integers(0, n_parts_);
        }
        if (sampled_) {
            Stream g(stream_key(seed_, SAMPLE, (uint64_t)t,
                                (uint64_t)(1 + replier)));
            return (int32_t)g.integers(0, (int64_t)1 << 31);
        }
        return -1;
    }

    // returns true if the reply was enqueued (false = dropped)
    bool enqueue_reply(int64_t t, int32_t replier, int32_t requester,
                       int32_t slot, int32_t pid, int64_t& sent,
                       int64_t& failed, int64_t& total_size)
    {
        Stream g(stream_key(seed_, DROP, (uint64_t)t, (uint64_t)(1 + replier)));
        double u = g.rnd();
        sent += 1;
        total_size += model_size_;
        if (u > drop_) {
            Stream gd(stream_key(seed_, DELAY, (uint64_t)t,
                                 (uint64_t)(1 + replier)));
            int64_t dly = delay_for(gd, model_size_);
            pend_push(t + dly, {requester, slot, -1, false, replier, pid});
            return true;
        }
        failed += 1;
        return false;
    }
};

py::dict NativeScheduler::next_round(int64_t r)
{
    auto t_start0 = std::chrono::steady_clock::now();
    const int64_t t0 = r * delta_, t1 = (r + 1) * delta_;
    int64_t sent = 0, failed = 0, total_size = 0;

    std::vector<int32_t> snap_nodes, snap_slots, snap_tptr{0};
    std::vector<int32_t> recv_nodes, recv_tptr{0};
    std::vector<int32_t> recv_nptr{0};
    std::vector<int32_t> del_slots, del_owners, reply_slots, del_pids;
    std::vector<int32_t> pull_nodes, pull_slots, pull_tptr{0};
    std::vector<int32_t> rep_nodes, rep_tptr{0}, rep_nptr{0};
    std::vector<int32_t> rep_slots, rep_owners, rep_pids;

    std::vector<double> online(n_);
    std::vector<int32_t> firing;
    firing.reserve(64);

    for (int64_t t = t0; t < t1; ++t) {
        std::vector<int32_t> freed;

        // --- firing set (ascending node id, like np.where)
        firing.clear();
        if (sync_) {
            firing = fire_buckets_[t % delta_];
        } else {
            for (int64_t i = 0; i < n_; ++i)
                if (t % deltas_[i] == 0) firing.push_back((int32_t)i);
        }

        // --- sends (gossipy/simul.py:393-407 equivalent)
        size_t n_f = firing.size();
        if (n_f) {
            Stream gp(stream_key(seed_, PEER, (uint64_t)t, 0));
            std::vector<int32_t> peers(n_f);
            if (indptr_.empty()) {
                for (size_t j = 0; j < n_f; ++j) {
                    int64_t draw = gp.integers(0, n_ - 1);
                    peers[j] = (int32_t)(draw + (draw >= firing[j] ? 1 : 0));
                }
            } else {
                // python draws ALL uniforms in one batch before indexing
                std::vector<double> u(n_f);
                for (size_t j = 0; j < n_f; ++j) u[j] = gp.rnd();
                for (size_t j = 0; j < n_f; ++j) {
                    int64_t s = indptr_[firing[j]];
                    int64_t deg = indptr_[firing[j] + 1] - s;
                    peers[j] = (int32_t)indices_[s + (int64_t)std::floor(
                                                          u[j] * (double)deg)];
                }
            }
            std::vector<double> drop_u(n_f, 1.0);
            if (drop_ > 0.0) {
                Stream gdrop(stream_key(seed_, DROP, (uint64_t)t, 0));
                for (size_t j = 0; j < n_f; ++j) drop_u[j] = gdrop.rnd();
            }
            int64_t msize = (proto_ == PULL) ? 1 : model_size_;
            std::vector<int64_t> delays(n_f);
            {
                Stream gdl(stream_key(seed_, DELAY, (uint64_t)t, 0));
                for (size_t j = 0; j < n_f; ++j) delays[j] = delay_for(gdl, msize);
            }
            std::vector<int32_t> pids(n_f, -1);
            if (n_parts_ > 0) {
                Stream gpt(stream_key(seed_, PART, (uint64_t)t, 0));
                for (size_t j = 0; j < n_f; ++j)
                    pids[j] = (int32_t)gpt.integers(0, n_parts_);
            } else if (sampled_) {
                Stream gsm(stream_key(seed_, SAMPLE, (uint64_t)t, 0));
                for (size_t j = 0; j < n_f; ++j)
                    pids[j] = (int32_t)gsm.integers(0, (int64_t)1 << 31);
            }
            for (size_t j = 0; j < n_f; ++j) {
                int32_t sender = firing[j], receiver = peers[j];
                bool is_pull = proto_ == PULL;
                int32_t slot = -1;
                if (!is_pull) {
                    slot = alloc_slot(sender, t);
                    snap_nodes.push_back(sender);
                    snap_slots.push_back(slot);
                }
                sent += 1;
                total_size += msize;
                if (drop_u[j] >= drop_) {
                    int64_t due = t + delays[j];
                    int32_t rf = (proto_ == PUSH_PULL) ? -2 : -1;
                    pend_push(due, {receiver, slot, rf, is_pull, sender, pids[j]});
                } else {
                    failed += 1;
                    if (slot >= 0) freed.push_back(slot);
                }
            }
        }
        snap_tptr.push_back((int32_t)snap_nodes.size());

        // --- deliveries due this tick (sub-phase B)
        if (online_ < 1.0) {
            Stream go(stream_key(seed_, ONLINE, (uint64_t)t, 0));
            for (int64_t i = 0; i < n_; ++i) online[i] = go.rnd();
        }
        // online_ >= 1.0: nobody is ever offline — the comparison below is
        // guarded, so the n-wide per-tick fill is skipped entirely
        std::vector<Msg> due = pend_take(t);
        // receiver -> (slot, rslot, sender) in first-appearance order
        std::vector<int32_t> order;
        ev_pool_.clear(); row_head_.clear(); row_tail_.clear();
        const int64_t st_main = 2 * t;
        auto row_append = [&](int64_t stamp, int32_t node, int32_t a,
                              int32_t b, int32_t c, int32_t d) {
            int32_t e = (int32_t)ev_pool_.size();
            ev_pool_.push_back({a, b, c, d, -1});
            if (row_stamp_[node] != stamp) {
                row_stamp_[node] = stamp;
                row_idx_[node] = (int32_t)order.size();
                order.push_back(node);
                row_head_.push_back(e);
                row_tail_.push_back(e);
            } else {
                ev_pool_[row_tail_[row_idx_[node]]].next = e;
                row_tail_[row_idx_[node]] = e;
            }
        };
        for (const Msg& m : due) {
            if (online_ < 1.0 && online[m.receiver] > online_) {
                failed += 1;
                if (m.slot >= 0) freed.push_back(m.slot);
                continue;
            }
            if (m.is_pull) {
                int32_t rslot = alloc_slot(m.receiver, t);
                pull_nodes.push_back(m.receiver);
                pull_slots.push_back(rslot);
                if (!enqueue_reply(t, m.receiver, m.sender, rslot,
                                   reply_pid(t, m.receiver), sent, failed,
                                   total_size))
                    freed.push_back(rslot);
                continue;
            }
            int32_t rslot = -1;
            if (m.reply_flag == -2) {
                rslot = alloc_slot(m.receiver, t);
                if (!enqueue_reply(t, m.receiver, m.sender, rslot,
                                   reply_pid(t, m.receiver), sent, failed,
                                   total_size))
                    freed.push_back(rslot);
            }
            row_append(st_main, m.receiver, m.slot, rslot, m.sender, m.pid);
            freed.push_back(m.slot);
        }
        for (size_t ri = 0; ri < order.size(); ++ri) {
            recv_nodes.push_back(order[ri]);
            for (int32_t e = row_head_[ri]; e >= 0; e = ev_pool_[e].next) {
                del_slots.push_back(ev_pool_[e].v0);
                reply_slots.push_back(ev_pool_[e].v1);
                del_owners.push_back(ev_pool_[e].v2);
                del_pids.push_back(ev_pool_[e].v3);
            }
            recv_nptr.push_back((int32_t)del_slots.size());
        }
        recv_tptr.push_back((int32_t)recv_nodes.size());
        pull_tptr.push_back((int32_t)pull_nodes.size());

        // --- sub-phase C: same-tick replies
        std::vector<Msg> rep_due = pend_take(t);
        std::vector<int32_t> rorder;
        ev_pool_.clear(); row_head_.clear(); row_tail_.clear();
        const int64_t st_rep = 2 * t + 1;
        auto rep_append = [&](int32_t node, int32_t a, int32_t b, int32_t c) {
            int32_t e = (int32_t)ev_pool_.size();
            ev_pool_.push_back({a, b, c, 0, -1});
            if (row_stamp_[node] != st_rep) {
                row_stamp_[node] = st_rep;
                row_idx_[node] = (int32_t)rorder.size();
                rorder.push_back(node);
                row_head_.push_back(e);
                row_tail_.push_back(e);
            } else {
                ev_pool_[row_tail_[row_idx_[node]]].next = e;
                row_tail_[row_idx_[node]] = e;
            }
        };
        for (const Msg& m : rep_due) {
            if (online_ < 1.0 && online[m.receiver] > online_) {
                failed += 1;
                freed.push_back(m.slot);
                continue;
            }
            rep_append(m.receiver, m.slot, m.sender, m.pid);
            freed.push_back(m.slot);
        }
        for (size_t ri = 0; ri < rorder.size(); ++ri) {
            rep_nodes.push_back(rorder[ri]);
            for (int32_t e = row_head_[ri]; e >= 0; e = ev_pool_[e].next) {
                rep_slots.push_back(ev_pool_[e].v0);
                rep_owners.push_back(ev_pool_[e].v1);
                rep_pids.push_back(ev_pool_[e].v2);
            }
            rep_nptr.push_back((int32_t)rep_slots.size());
        }
        rep_tptr.push_back((int32_t)rep_nodes.size());

        for (int32_t s : freed) reuse_q_.emplace_back(t, s);
    }

    auto t_mid0 = std::chrono::steady_clock::now();
    py::dict out;
    auto arr = [](std::vector<int32_t>& v) {
        auto a = py::array_t<int32_t>((py::ssize_t)v.size());
        std::copy(v.begin(), v.end(), a.mutable_data());
        return a;
    };
    out["snap_tptr"] = arr(snap_tptr);
    out["recv_tptr"] = arr(recv_tptr);
    out["pull_tptr"] = arr(pull_tptr);
    out["rep_tptr"] = arr(rep_tptr);
    if (!lean_) {
        out["snap_nodes"] = arr(snap_nodes);
        out["snap_slots"] = arr(snap_slots);
        out["recv_nodes"] = arr(recv_nodes);
        out["recv_nptr"] = arr(recv_nptr);
        out["del_slots"] = arr(del_slots);
        out["del_owners"] = arr(del_owners);
        out["reply_slots"] = arr(reply_slots);
        out["pull_nodes"] = arr(pull_nodes);
        out["pull_slots"] = arr(pull_slots);
        out["rep_nodes"] = arr(rep_nodes);
        out["rep_nptr"] = arr(rep_nptr);
        out["rep_slots"] = arr(rep_slots);
        out["rep_owners"] = arr(rep_owners);
        out["del_pids"] = arr(del_pids);
        out["rep_pids"] = arr(rep_pids);
    }

    out["sent"] = sent;
    out["failed"] = failed;
    out["total_size"] = total_size;
    out["n_slots"] = next_slot_;
    {
        if (!lean_) {
            auto mb = compute_merge_bounds(
                n_, next_slot_, snap_nodes, snap_slots, snap_tptr, recv_nodes,
                recv_nptr, recv_tptr, del_slots, reply_slots, pull_nodes,
                pull_slots, pull_tptr, rep_nodes, rep_nptr, rep_tptr,
                rep_slots);
            out["merge_bounds"] = arr(mb);
        }
        out["packed"] = pack_round(
            snap_nodes, snap_slots, snap_tptr, recv_nodes, recv_nptr,
            recv_tptr, del_slots, reply_slots, del_pids, pull_nodes,
            pull_slots, pull_tptr, rep_nodes, rep_nptr, rep_tptr, rep_slots,
            rep_pids, del_owners, rep_owners, n_, next_slot_);
    }
    if (sampling_eval_ > 0) {
        Stream g(stream_key(seed_, EVAL, (uint64_t)(t1 - 1), 0));
        int64_t k = std::max<int64_t>((int64_t)(n_ * sampling_eval_), 1);
        auto ev = py::array_t<int64_t>(k);
        for (int64_t i = 0; i < k; ++i)
            ev.mutable_data()[i] = g.integers(0, n_);
        out["eval_nodes"] = ev;
    } else {
        out["eval_nodes"] = py::none();
    }
    if (getenv("GOSSIPY_SCHED_TIME")) {
        static double t_loop = 0, t_emit = 0;
        auto t_end0 = std::chrono::steady_clock::now();
        t_loop += std::chrono::duration<double>(t_mid0 - t_start0).count();
        t_emit += std::chrono::duration<double>(t_end0 - t_mid0).count();
        fprintf(stderr, "[sched] loop=%.3fms emit=%.3fms (cum)\n",
                t_loop * 1000, t_emit * 1000);
    }
    return out;
}


// ---------------------------------------------------------------------------
// Tokenized (flow-controlled) scheduler — C++ replica of
// engine/schedule.py TokenizedScheduler (bit-exact; equivalence enforced by
// tests/test_native_sched.py). Emits flat arrays whose "tick" groups are
// the python scheduler's phases: one group per delivery WAVE (reactive
// bursts with zero delay cascade within a tick) plus one group per
// same-tick reply batch. Constant utility only (the reference experiments
// use utility == 1, main_hegedus_2021.py:57); callables stay in python.
// ---------------------------------------------------------------------------

enum AccountKind {
    ACC_PURELY_PROACTIVE = 0,
    ACC_PURELY_REACTIVE = 1,
    ACC_SIMPLE = 2,
    ACC_GENERALIZED = 3,
    ACC_RANDOMIZED = 4
};

struct Account {  // matches gossipy_amd/flow_control.py
    int kind;
    double capacity;   // C
    double reactivity; // A
    double k;          // purely-reactive multiplier
    int64_t n_tokens = 0;

    void add(int64_t n) { n_tokens += n; }
    void sub(int64_t n) { n_tokens = std::max<int64_t>(0, n_tokens - n); }

    double proactive() const
    {
        switch (kind) {
        case ACC_PURELY_PROACTIVE: return 1.0;
        case ACC_PURELY_REACTIVE: return 0.0;
        case ACC_SIMPLE: return n_tokens >= capacity ? 1.0 : 0.0;
        case ACC_GENERALIZED: return n_tokens >= capacity ? 1.0 : 0.0;
        default:  // randomized: linear ramp (flow_control.py:116-123)
            if ((double)n_tokens < reactivity - 1) return 0.0;
            if ((double)n_tokens <= capacity)
                return ((double)n_tokens - reactivity + 1) /
                       (capacity - reactivity + 1);
            return 1.0;
        }
    }

    int64_t reactive(int64_t utility, double u) const
    {
        switch (kind) {
        case ACC_PURELY_PROACTIVE: return 0;
        case ACC_PURELY_REACTIVE: return (int64_t)(utility * k);
        case ACC_SIMPLE: return n_tokens > 0 ? 1 : 0;
        case ACC_GENERALIZED: {
            double num = reactivity + (double)n_tokens - 1;
            return (int64_t)(utility > 0 ? num / reactivity
                                         : num / (2 * reactivity));
        }
        default: {  // randomized rounding with the tape draw
            if (utility <= 0) return 0;
            double r = (double)n_tokens / reactivity;
            double frac = r - std::floor(r);
            return (int64_t)r + (u < frac ? 1 : 0);
        }
        }
    }
};

class NativeTokenizedScheduler : public NativeScheduler {
public:
    NativeTokenizedScheduler(int64_t n_nodes, int64_t delta, int proto,
                             int64_t model_size, double drop_prob,
                             double online_prob, int delay_kind, int64_t dmin,
                             int64_t dmax, double timexunit, int64_t overhead,
                             bool sync, double sampling_eval, uint64_t seed,
                             py::object peers_indptr, py::object peers_indices,
                             int64_t n_parts, bool sampled, int account_kind,
                             double acc_C, double acc_A, double acc_k,
                             int64_t utility)
        : NativeScheduler(n_nodes, delta, proto, model_size, drop_prob,
                          online_prob, delay_kind, dmin, dmax, timexunit,
                          overhead, sync, sampling_eval, seed, peers_indptr,
                          peers_indices, n_parts, sampled),
          utility_(utility)
    {
        Account proto_acc{account_kind, acc_C, acc_A, acc_k, 0};
        accounts_.assign(n_, proto_acc);
    }

    py::dict next_round(int64_t r);

    py::list token_balances() const
    {
        py::list out;
        for (const auto& a : accounts_) out.append(a.n_tokens);
        return out;
    }

private:
    std::vector<Account> accounts_;
    int64_t utility_;

    int32_t reply_extra(int64_t t, int32_t replier)
    {
        if (n_parts_ > 0) {
            Stream g(stream_key(seed_, PART, (uint64_t)t,
                                (uint64_t)(1 + replier)));
            return (int32_t)g.integers(0, n_parts_);
        }
        if (sampled_) {
            Stream g(stream_key(seed_, SAMPLE, (uint64_t)t,
                                (uint64_t)(1 + replier)));
            return (int32_t)g.integers(0, (int64_t)1 << 31);
        }
        return -1;
    }

    int32_t burst_peer(int32_t node, Stream& gp)
    {
        if (indptr_.empty()) {
            int64_t draw = gp.integers(0, n_ - 1);
            return (int32_t)(draw + (draw >= node ? 1 : 0));
        }
        int64_t s = indptr_[node];
        int64_t deg = indptr_[node + 1] - s;
        return (int32_t)indices_[s + (int64_t)std::floor(gp.rnd() * (double)deg)];
    }

    int64_t burst_delay(Stream& gdl)
    {
        switch (dkind_) {
        case D_CONST: return dmin_;
        case D_UNIFORM: return gdl.integers(dmin_, dmax_ + 1);
        default: return (int64_t)(timexunit_ * (double)model_size_) + overhead_;
        }
    }
};

py::dict NativeTokenizedScheduler::next_round(int64_t r)
{
    const int64_t t0 = r * delta_, t1 = (r + 1) * delta_;
    int64_t sent = 0, failed = 0, total_size = 0;

    std::vector<int32_t> snap_nodes, snap_slots, snap_tptr{0};
    std::vector<int32_t> recv_nodes, recv_tptr{0}, recv_nptr{0};
    std::vector<int32_t> del_slots, del_owners, reply_slots, del_pids;
    std::vector<int32_t> pull_nodes, pull_slots, pull_tptr{0};
    std::vector<int32_t> rep_nodes, rep_tptr{0}, rep_nptr{0};
    std::vector<int32_t> rep_slots, rep_owners, rep_pids;
    std::vector<double> online(n_);

    // a "group" boundary closes every per-kind tick pointer
    auto close_group = [&]() {
        snap_tptr.push_back((int32_t)snap_nodes.size());
        recv_tptr.push_back((int32_t)recv_nodes.size());
        pull_tptr.push_back((int32_t)pull_nodes.size());
        rep_tptr.push_back((int32_t)rep_nodes.size());
    };

    std::vector<int32_t> firing;
    for (int64_t t = t0; t < t1; ++t) {
        std::vector<int32_t> freed;
        firing.clear();
        if (sync_) {
            firing = fire_buckets_[t % delta_];
        } else {
            for (int64_t i = 0; i < n_; ++i)
                if (t % deltas_[i] == 0) firing.push_back((int32_t)i);
        }

        // --- proactive-gated sends
        std::vector<int32_t> wave_snap_nodes, wave_snap_slots;
        size_t n_f = firing.size();
        if (n_f) {
            Stream gpro(stream_key(seed_, TOKEN, (uint64_t)t, 0));
            std::vector<double> pro_u(n_f);
            for (size_t j = 0; j < n_f; ++j) pro_u[j] = gpro.rnd();
            std::vector<int32_t> senders;
            for (size_t j = 0; j < n_f; ++j) {
                if (pro_u[j] < accounts_[firing[j]].proactive())
                    senders.push_back(firing[j]);
                else
                    accounts_[firing[j]].add(1);
            }
            size_t n_s = senders.size();
            if (n_s) {
                // peers (python _peers_of draw order)
                Stream gp(stream_key(seed_, PEER, (uint64_t)t, 0));
                std::vector<int32_t> peers(n_s);
                if (indptr_.empty()) {
                    for (size_t j = 0; j < n_s; ++j) {
                        int64_t draw = gp.integers(0, n_ - 1);
                        peers[j] = (int32_t)(draw + (draw >= senders[j] ? 1 : 0));
                    }
                } else {
                    std::vector<double> u(n_s);
                    for (size_t j = 0; j < n_s; ++j) u[j] = gp.rnd();
                    for (size_t j = 0; j < n_s; ++j) {
                        int64_t st = indptr_[senders[j]];
                        int64_t deg = indptr_[senders[j] + 1] - st;
                        peers[j] = (int32_t)indices_[st + (int64_t)std::floor(
                                                             u[j] * (double)deg)];
                    }
                }
                Stream gdrop(stream_key(seed_, DROP, (uint64_t)t, 0));
                std::vector<double> drop_u(n_s);
                for (size_t j = 0; j < n_s; ++j) drop_u[j] = gdrop.rnd();
                int64_t msize =
                    (proto_ == PULL) ? 1 : model_size_;
                std::vector<int64_t> delays(n_s);
                {
                    Stream gdl(stream_key(seed_, DELAY, (uint64_t)t, 0));
                    for (size_t j = 0; j < n_s; ++j)
                        delays[j] = delay_for(gdl, msize);
                }
                std::vector<int32_t> pids(n_s, -1);
                if (n_parts_ > 0) {
                    Stream gpt(stream_key(seed_, PART, (uint64_t)t, 0));
                    for (size_t j = 0; j < n_s; ++j)
                        pids[j] = (int32_t)gpt.integers(0, n_parts_);
                } else if (sampled_) {
                    Stream gsm(stream_key(seed_, SAMPLE, (uint64_t)t, 0));
                    for (size_t j = 0; j < n_s; ++j)
                        pids[j] = (int32_t)gsm.integers(0, (int64_t)1 << 31);
                }
                for (size_t j = 0; j < n_s; ++j) {
                    int32_t sender = senders[j], receiver = peers[j];
                    bool is_pull = proto_ == PULL;
                    int32_t slot = -1;
                    if (!is_pull) {
                        slot = alloc_slot(sender, t);
                        wave_snap_nodes.push_back(sender);
                        wave_snap_slots.push_back(slot);
                    }
                    sent += 1;
                    total_size += msize;
                    if (drop_u[j] >= drop_) {
                        int32_t rf = (proto_ == PUSH_PULL) ? -2 : -1;
                        pend_push(t + delays[j],
                            {receiver, slot, rf, is_pull, sender, pids[j]});
                    } else {
                        failed += 1;
                        if (slot >= 0) freed.push_back(slot);
                    }
                }
            }
        }

        if (online_ < 1.0) {
            Stream go(stream_key(seed_, ONLINE, (uint64_t)t, 0));
            for (int64_t i = 0; i < n_; ++i) online[i] = go.rnd();
        }
        // online_ >= 1.0: nobody is ever offline — the comparison below is
        // guarded, so the n-wide per-tick fill is skipped entirely

        // --- delivery waves with reactive bursts
        std::vector<Msg> wave_due = pend_take(t);
        // per-(purpose, node) streams for this tick's bursts
        std::map<std::pair<int, int32_t>, Stream> tstreams;
        auto tick_stream = [&](int purpose, int32_t node) -> Stream& {
            auto key = std::make_pair(purpose, node);
            auto it = tstreams.find(key);
            if (it == tstreams.end())
                it = tstreams
                         .emplace(key, Stream(stream_key(seed_, purpose,
                                                         (uint64_t)t,
                                                         (uint64_t)(1 + node))))
                         .first;
            return it->second;
        };

        while (!wave_due.empty() || !wave_snap_nodes.empty()) {
            // group start: this wave's snapshots
            for (size_t j = 0; j < wave_snap_nodes.size(); ++j) {
                snap_nodes.push_back(wave_snap_nodes[j]);
                snap_slots.push_back(wave_snap_slots[j]);
            }
            wave_snap_nodes.clear();
            wave_snap_slots.clear();

            std::vector<Msg> next_due;
            std::vector<int32_t> order;
            std::unordered_map<int32_t, std::vector<std::array<int32_t, 4>>> rmap;
            for (const Msg& m : wave_due) {
                if (online[m.receiver] > online_) {
                    failed += 1;
                    if (m.slot >= 0) freed.push_back(m.slot);
                    continue;
                }
                if (m.is_pull) {
                    int32_t rslot = alloc_slot(m.receiver, t);
                    pull_nodes.push_back(m.receiver);
                    pull_slots.push_back(rslot);
                    if (!enqueue_reply(t, m.receiver, m.sender, rslot,
                                       reply_extra(t, m.receiver), sent,
                                       failed, total_size))
                        freed.push_back(rslot);
                    continue;
                }
                int32_t rslot = -1;
                if (m.reply_flag == -2) {
                    rslot = alloc_slot(m.receiver, t);
                    if (!enqueue_reply(t, m.receiver, m.sender, rslot,
                                       reply_extra(t, m.receiver), sent,
                                       failed, total_size))
                        freed.push_back(rslot);
                }
                auto f = rmap.find(m.receiver);
                if (f == rmap.end()) {
                    order.push_back(m.receiver);
                    rmap[m.receiver] = {{m.slot, rslot, m.pid, m.sender}};
                } else {
                    f->second.push_back({m.slot, rslot, m.pid, m.sender});
                }
                freed.push_back(m.slot);
                // reactive burst on reply-free deliveries (flag -1; REPLY
                // payloads carry -3 and never react)
                if (m.reply_flag == -1) {
                    double ru = tick_stream(TOKEN, m.receiver).rnd();
                    int64_t reaction =
                        accounts_[m.receiver].reactive(utility_, ru);
                    if (reaction <= 0) continue;
                    accounts_[m.receiver].sub(reaction);
                    Stream& gp = tick_stream(PEER, m.receiver);
                    Stream& gd = tick_stream(DROP, m.receiver);
                    Stream& gdl = tick_stream(DELAY, m.receiver);
                    for (int64_t b = 0; b < reaction; ++b) {
                        int32_t peer = burst_peer(m.receiver, gp);
                        int32_t bslot = alloc_slot(m.receiver, t);
                        wave_snap_nodes.push_back(m.receiver);
                        wave_snap_slots.push_back(bslot);
                        int32_t bpid = -1;
                        if (n_parts_ > 0)
                            bpid = (int32_t)tick_stream(PART, m.receiver)
                                       .integers(0, n_parts_);
                        else if (sampled_)
                            bpid = (int32_t)tick_stream(SAMPLE, m.receiver)
                                       .integers(0, (int64_t)1 << 31);
                        sent += 1;
                        total_size += model_size_;
                        if (gd.rnd() >= drop_) {
                            int64_t dly = burst_delay(gdl);
                            int32_t rf = (proto_ == PUSH_PULL) ? -2 : -1;
                            Msg bm{peer, bslot, rf, false, m.receiver, bpid};
                            if (dly == 0)
                                next_due.push_back(bm);
                            else
                                pend_push(t + dly, bm);
                        } else {
                            failed += 1;
                            freed.push_back(bslot);
                        }
                    }
                }
            }
            for (int32_t rn : order) {
                recv_nodes.push_back(rn);
                for (auto& p : rmap[rn]) {
                    del_slots.push_back(p[0]);
                    reply_slots.push_back(p[1]);
                    del_pids.push_back(p[2]);
                    del_owners.push_back(p[3]);
                }
                recv_nptr.push_back((int32_t)del_slots.size());
            }
            close_group();
            wave_due = std::move(next_due);
        }

        // --- same-tick replies: their own group
        std::vector<Msg> rep_due = pend_take(t);
        if (!rep_due.empty()) {
            std::vector<int32_t> rorder;
            std::unordered_map<int32_t, std::vector<std::array<int32_t, 3>>> rrmap;
            for (const Msg& m : rep_due) {
                if (online[m.receiver] > online_) {
                    failed += 1;
                    freed.push_back(m.slot);
                    continue;
                }
                auto f = rrmap.find(m.receiver);
                if (f == rrmap.end()) {
                    rorder.push_back(m.receiver);
                    rrmap[m.receiver] = {{m.slot, m.sender, m.pid}};
                } else {
                    f->second.push_back({m.slot, m.sender, m.pid});
                }
                freed.push_back(m.slot);
            }
            for (int32_t rn : rorder) {
                rep_nodes.push_back(rn);
                for (auto& p : rrmap[rn]) {
                    rep_slots.push_back(p[0]);
                    rep_owners.push_back(p[1]);
                    rep_pids.push_back(p[2]);
                }
                rep_nptr.push_back((int32_t)rep_slots.size());
            }
            close_group();
        }

        for (int32_t sfree : freed) reuse_q_.emplace_back(t, sfree);
    }

    py::dict out;
    auto arr = [](std::vector<int32_t>& v) {
        auto a = py::array_t<int32_t>((py::ssize_t)v.size());
        std::copy(v.begin(), v.end(), a.mutable_data());
        return a;
    };
    out["snap_tptr"] = arr(snap_tptr);
    out["recv_tptr"] = arr(recv_tptr);
    out["pull_tptr"] = arr(pull_tptr);
    out["rep_tptr"] = arr(rep_tptr);
    if (!lean_) {
        out["snap_nodes"] = arr(snap_nodes);
        out["snap_slots"] = arr(snap_slots);
        out["recv_nodes"] = arr(recv_nodes);
        out["recv_nptr"] = arr(recv_nptr);
        out["del_slots"] = arr(del_slots);
        out["del_owners"] = arr(del_owners);
        out["reply_slots"] = arr(reply_slots);
        out["del_pids"] = arr(del_pids);
        out["pull_nodes"] = arr(pull_nodes);
        out["pull_slots"] = arr(pull_slots);
        out["rep_nodes"] = arr(rep_nodes);
        out["rep_nptr"] = arr(rep_nptr);
        out["rep_slots"] = arr(rep_slots);
        out["rep_owners"] = arr(rep_owners);
        out["rep_pids"] = arr(rep_pids);
    }

    out["sent"] = sent;
    out["failed"] = failed;
    out["total_size"] = total_size;
    out["n_slots"] = next_slot_;
    {
        if (!lean_) {
            auto mb = compute_merge_bounds(
                n_, next_slot_, snap_nodes, snap_slots, snap_tptr, recv_nodes,
                recv_nptr, recv_tptr, del_slots, reply_slots, pull_nodes,
                pull_slots, pull_tptr, rep_nodes, rep_nptr, rep_tptr,
                rep_slots);
            out["merge_bounds"] = arr(mb);
        }
        out["packed"] = pack_round(
            snap_nodes, snap_slots, snap_tptr, recv_nodes, recv_nptr,
            recv_tptr, del_slots, reply_slots, del_pids, pull_nodes,
            pull_slots, pull_tptr, rep_nodes, rep_nptr, rep_tptr, rep_slots,
            rep_pids, del_owners, rep_owners, n_, next_slot_);
    }
    if (sampling_eval_ > 0) {
        Stream g(stream_key(seed_, EVAL, (uint64_t)(t1 - 1), 0));
        int64_t k = std::max<int64_t>((int64_t)(n_ * sampling_eval_), 1);
        auto ev = py::array_t<int64_t>(k);
        for (int64_t i = 0; i < k; ++i)
            ev.mutable_data()[i] = g.integers(0, n_);
        out["eval_nodes"] = ev;
    } else {
        out["eval_nodes"] = py::none();
    }
    return out;
}

PYBIND11_MODULE(_gossip_sched, m)
{
    py::class_<NativeScheduler>(m, "NativeScheduler")
        .def(py::init<int64_t, int64_t, int, int64_t, double, double, int,
                      int64_t, int64_t, double, int64_t, bool, double,
                      uint64_t, py::object, py::object, int64_t, bool>(),
             py::arg("n_nodes"), py::arg("delta"), py::arg("proto"),
             py::arg("model_size"), py::arg("drop_prob"), py::arg("online_prob"),
             py::arg("delay_kind"), py::arg("dmin"), py::arg("dmax"),
             py::arg("timexunit"), py::arg("overhead"), py::arg("sync"),
             py::arg("sampling_eval"), py::arg("seed"),
             py::arg("peers_indptr"), py::arg("peers_indices"),
             py::arg("n_parts") = 0, py::arg("sampled") = false)
        .def("next_round", &NativeScheduler::next_round)
        .def("set_lean", &NativeScheduler::set_lean);
    py::class_<NativeTokenizedScheduler>(m, "NativeTokenizedScheduler")
        .def(py::init<int64_t, int64_t, int, int64_t, double, double, int,
                      int64_t, int64_t, double, int64_t, bool, double,
                      uint64_t, py::object, py::object, int64_t, bool, int,
                      double, double, double, int64_t>())
        .def("next_round", &NativeTokenizedScheduler::next_round)
        .def("set_lean", &NativeTokenizedScheduler::set_lean)
        .def("token_balances", &NativeTokenizedScheduler::token_balances);
}
