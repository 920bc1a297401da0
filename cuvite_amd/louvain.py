"""Multi-phase distributed Louvain orchestration.

Structure mirrors the reference: an outer phase loop
(local-moving until convergence, then graph coarsening) around an inner
iteration loop (halo fill -> local move -> community update -> remote update
-> modularity -> convergence test). Variants: plain, early-termination
(freeze / probabilistic, ref louvain.cpp:7-424), coloring-ordered and
vertex-ordered moves (ref louvain.cpp:756-2101), threshold cycling
(main.cpp:225-239).

All state lives on the compute device; per-iteration host work is zero on the
GPU path (the reference rebuilds its dense community remap on the host every
iteration, louvain_cuda.cu:2260-2378 — here remote community ids are interned
once into a phase-persistent on-device universe and only CHANGED labels are
remapped per iteration, with no sort/unique in the loop; see
PhaseState.densify).
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from .graph import DistGraph
from .halo import (build_halo, exchange_ghost_labels, fetch_comm_info_lists,
                   push_remote_deltas)
from .local_move import MoveInputs, local_move_torch, modularity_parts
from .ops import scatter_add_
from .parallel import Comm
from .utils.timers import Timers

TERMINATION_PHASE_COUNT = 200   # ref utils.hpp:17-19
MAX_TOTAL_ITERS = 10000         # ref main.cpp:486-494
ET_CUTOFF = 90                  # ref louvain.hpp:76 (absolute count)
P_CUTOFF = 0.02                 # ref louvain.hpp:78


@dataclass
class LouvainConfig:
    threshold: float = 1.0e-6
    threshold_scaling: bool = False      # -i
    one_phase: bool = False              # -p
    early_term: int = 0                  # -t 1..4 (0 = off)
    et_delta: float = 1.0                # -a (alpha for -t 2/4)
    coloring: bool = False               # -c
    ordering: bool = False               # -d
    max_colors: int = 8                  # -c/-d arg (nHash*2)
    backend: str = "auto"                # "auto" | "torch" | "hip"
    max_phases: int = TERMINATION_PHASE_COUNT
    max_iters_per_phase: int = 10000
    verbose: bool = False


@dataclass
class LouvainResult:
    modularity: float
    communities: torch.Tensor   # final community id per ORIGINAL local vertex
    phases: int
    total_iters: int
    times: dict = field(default_factory=dict)
    modularity_per_level: list = field(default_factory=list)
    levels: list = field(default_factory=list)  # per-phase {ne_global, iters,
    #   modularity, cluster_s, rebuild_s} — feeds the reference TEPS
    #   definition (teps += ne * iters per phase, main.cpp:448)

    @property
    def teps_numerator(self) -> float:
        return sum(lv["ne_global"] * lv["iters"] for lv in self.levels)


def _threshold_for_phase(cfg: LouvainConfig, short_phase: int) -> float:
    """Threshold cycling (ref main.cpp:225-239)."""
    if not cfg.threshold_scaling or cfg.one_phase:
        return 1.0e-6
    sp = short_phase % 13
    if sp <= 2:
        return 1.0e-3
    if sp <= 6:
        return 1.0e-4
    if sp <= 9:
        return 1.0e-5
    return 1.0e-6


def _pick_move_fn(cfg: LouvainConfig, device: torch.device):
    if cfg.backend == "torch":
        return local_move_torch
    if cfg.backend in ("auto", "hip") and device.type == "cuda":
        from . import ops
        return ops.local_move  # raises if the HIP extension is missing
    if cfg.backend == "hip":
        raise RuntimeError("backend='hip' requires a CUDA/HIP device")
    return local_move_torch


class PhaseState:
    """Mutable per-phase state for one rank (all tensors on dg's device)."""

    def __init__(self, dg: DistGraph, comm: Comm, halo=None):
        dev = dg.g.device
        self.dg = dg
        self.comm = comm
        self.halo = halo if halo is not None else build_halo(dg, comm)
        nv = dg.nv
        W = dg.g.weights.dtype
        self.v_degree = dg.local_degree_sum()                       # W [nv]
        # community aggregates, indexed by (gid - base) for owned comms
        self.local_size = torch.ones(nv, dtype=torch.int64, device=dev)
        self.local_degree = self.v_degree.clone()
        # global labels (ref distInitComm: singletons)
        arange = torch.arange(dg.base, dg.bound, device=dev)
        self.curr_comm = arange.clone()
        self.past_comm = arange.clone()
        # constant = 1/(2m) (ref distCalcConstantForSecondTerm; guard the
        # edgeless-graph case the reference would divide by zero on)
        tw = float(self.v_degree.to(torch.float64).sum())
        tw = comm.allreduce_scalar(tw)
        self.constant = 1.0 / tw if tw > 0.0 else 0.0
        # ET state
        self.active = torch.ones(nv, dtype=torch.bool, device=dev)
        self.stable_count = torch.zeros(nv, dtype=torch.int16, device=dev)
        self.move_prob = torch.ones(nv, dtype=torch.float32, device=dev)
        # phase-persistent dense community space (see densify)
        self._lgids: Optional[torch.Tensor] = None
        self._parts_dev: Optional[torch.Tensor] = None
        self._runiv: Optional[torch.Tensor] = None
        self._prev_labels: Optional[torch.Tensor] = None

    def _local_gids(self) -> torch.Tensor:
        if self._lgids is None:
            self._lgids = torch.arange(self.dg.base, self.dg.bound,
                                       device=self.dg.g.device)
        return self._lgids

    def _univ_lookup(self, rem: torch.Tensor):
        """Position in the (append-ordered) universe for each gid in `rem`,
        plus a found mask (positions are garbage where not found)."""
        K = self._runiv_sorted.numel()
        if K == 0:
            z = torch.zeros(rem.numel(), dtype=torch.bool, device=rem.device)
            return torch.zeros(rem.numel(), dtype=torch.int64,
                               device=rem.device), z
        idx = torch.searchsorted(self._runiv_sorted, rem).clamp(max=K - 1)
        found = self._runiv_sorted[idx] == rem
        return self._runiv_perm[idx], found

    def _univ_append(self, new: torch.Tensor):
        """Append new remote community gids to the universe (dense ids are
        append-ordered so existing ids stay stable); rebuild the sorted
        lookup index (only runs on growth — rare after iteration 1)."""
        dev = new.device
        if self._parts_dev is None:
            self._parts_dev = self.dg.partition.parts.to(dev)
        owner = torch.searchsorted(self._parts_dev, new, right=True) - 1
        self._runiv = torch.cat([self._runiv, new])
        self._univ_owner = torch.cat([self._univ_owner, owner])
        self._univ_size = torch.cat([
            self._univ_size,
            torch.zeros(new.numel(), dtype=torch.int64, device=dev)])
        self._univ_degree = torch.cat([
            self._univ_degree,
            torch.zeros(new.numel(), dtype=self.local_degree.dtype,
                        device=dev)])
        self._runiv_sorted, self._runiv_perm = torch.sort(self._runiv)

    def densify(self, ghost_comm: torch.Tensor):
        """Map global labels to the dense community space.
        Returns (curr_dense [nv+ng] int32, remote_gids (append-ordered
        universe), comm_size, comm_degree, comm_gid) with remote info
        fetched from owners for every REFERENCED remote community.

        Phase-persistent design (SURVEY section 7 "biggest systems win"; the
        reference rebuilds its dense remap on the host every iteration,
        louvain_cuda.cu:2260-2378): remote community gids are interned into
        an append-only universe, so per-iteration work is (a) an O(n)
        changed-label compare against the previous iteration's labels,
        (b) a searchsorted lookup for the changed ones only, and (c) an O(n)
        scatter for the referenced-set mask — NO per-iteration torch.unique /
        sort over nv+ng labels. Universe entries that are no longer
        referenced keep stale aggregates; the move kernels only ever read
        communities adjacent to some vertex, which are referenced and fresh
        this iteration."""
        dg, dev = self.dg, self.dg.g.device
        base, bound, nv = dg.base, dg.bound, dg.nv
        if self.comm.world == 1:
            # single-rank fast path: every label is local
            dense = (self.curr_comm - base).to(torch.int32)
            empty = torch.empty(0, dtype=torch.int64, device=dev)
            return (dense, empty, self.local_size, self.local_degree,
                    self._local_gids())
        all_labels = torch.cat([self.curr_comm, ghost_comm])
        n = all_labels.numel()
        if self._runiv is None:
            i64 = dict(dtype=torch.int64, device=dev)
            self._runiv = torch.empty(0, **i64)
            self._runiv_sorted = self._runiv
            self._runiv_perm = torch.empty(0, **i64)
            self._univ_owner = torch.empty(0, **i64)
            self._univ_size = torch.empty(0, **i64)
            self._univ_degree = torch.empty(
                0, dtype=self.local_degree.dtype, device=dev)
            self._dense = torch.empty(n, dtype=torch.int32, device=dev)

        if self._prev_labels is not None and self._prev_labels.numel() == n:
            changed = (all_labels != self._prev_labels).nonzero(
                as_tuple=True)[0]
            lab_c = all_labels[changed]
        else:
            changed = None
            lab_c = all_labels

        if lab_c.numel():
            is_local = (lab_c >= base) & (lab_c < bound)
            rem = lab_c[~is_local]
            pos, found = self._univ_lookup(rem)
            if not bool(found.all()):
                self._univ_append(torch.unique(rem[~found]))
                pos, _ = self._univ_lookup(rem)
            dense_c = torch.empty(lab_c.numel(), dtype=torch.int32,
                                  device=dev)
            dense_c[is_local] = (lab_c[is_local] - base).to(torch.int32)
            if rem.numel():
                dense_c[~is_local] = (nv + pos).to(torch.int32)
            if changed is None:
                self._dense = dense_c
            else:
                self._dense[changed] = dense_c
        self._prev_labels = all_labels

        # referenced remote communities this iteration (mask in universe
        # order), then one request/reply round with their owners
        K = self._runiv.numel()
        ref_mask = torch.zeros(K, dtype=torch.bool, device=dev)
        d64 = self._dense.to(torch.int64)
        rr = d64[d64 >= nv] - nv
        ref_mask[rr] = True
        world = self.comm.world
        reqs, positions = [], []
        for p in range(world):
            sel = (ref_mask & (self._univ_owner == p)).nonzero(
                as_tuple=True)[0]
            reqs.append(self._runiv[sel])
            positions.append(sel)
        sizes_l, degrees_l = fetch_comm_info_lists(
            self.halo, reqs, self.local_size, self.local_degree)
        for p in range(world):
            if positions[p].numel():
                self._univ_size[positions[p]] = sizes_l[p]
                self._univ_degree[positions[p]] = degrees_l[p]
        comm_size = torch.cat([self.local_size, self._univ_size])
        comm_degree = torch.cat([self.local_degree, self._univ_degree])
        comm_gid = torch.cat([self._local_gids(), self._runiv])
        return self._dense, self._runiv, comm_size, comm_degree, comm_gid

    def apply_moves(self, target_gid: torch.Tensor, remote_gids: torch.Tensor):
        """Apply community size/degree deltas for vertices that moved
        (ref 4-case update, louvain.cpp:2308-2376 + updateRemoteCommunities)."""
        dg = self.dg
        base, bound = dg.base, dg.bound
        if getattr(self, "use_hip", False) and target_gid.is_cuda:
            from . import ops
            if self.comm.world == 1:
                # fresh recount beats the delta update when most vertices
                # move (2 atomics/vertex vs 4 per moved vertex; rocprof
                # 25 -> ~12 ms at s26's oscillating sweeps)
                ops._require().recount_(target_gid, self.v_degree, base,
                                        self.local_size, self.local_degree)
                return
            # fused local delta update, no host sync / compaction kernels
            ops.apply_deltas_(target_gid, self.curr_comm, self.v_degree,
                              base, bound, self.local_size,
                              self.local_degree)
            moved = target_gid != self.curr_comm
            src = self.curr_comm[moved]
            dst = target_gid[moved]
            vdeg = self.v_degree[moved]
            gids = torch.cat([src, dst])
            dsize = torch.cat([-torch.ones_like(src), torch.ones_like(dst)])
            ddeg = torch.cat([-vdeg, vdeg])
            rem = (gids < base) | (gids >= bound)
            push_remote_deltas(self.halo, gids[rem], dsize[rem], ddeg[rem],
                               self.local_size, self.local_degree)
            return
        moved = target_gid != self.curr_comm
        if not bool(moved.any()):
            # still must participate in the collective delta push
            if self.comm.world > 1:
                empty_i = torch.empty(0, dtype=torch.int64, device=target_gid.device)
                empty_w = torch.empty(0, dtype=self.local_degree.dtype,
                                      device=target_gid.device)
                push_remote_deltas(self.halo, empty_i, empty_i, empty_w,
                                   self.local_size, self.local_degree)
            return
        src = self.curr_comm[moved]
        dst = target_gid[moved]
        vdeg = self.v_degree[moved]
        gids = torch.cat([src, dst])
        dsize = torch.cat([-torch.ones_like(src), torch.ones_like(dst)])
        ddeg = torch.cat([-vdeg, vdeg])
        is_local = (gids >= base) & (gids < bound)
        li = gids[is_local] - base
        self.local_size.index_add_(0, li, dsize[is_local])
        scatter_add_(self.local_degree, li, ddeg[is_local])
        if self.comm.world > 1:
            push_remote_deltas(self.halo, gids[~is_local], dsize[~is_local],
                               ddeg[~is_local], self.local_size, self.local_degree)
        else:
            assert bool(is_local.all())


def _et_update(state: PhaseState, cfg: LouvainConfig, target: torch.Tensor,
               rng: torch.Generator) -> Optional[int]:
    """Early-termination bookkeeping after an iteration. Returns the global
    frozen count for -t 3/4 (ref louvain.cpp:115-120, 172-182, 378-395)."""
    if cfg.early_term in (1, 3):
        stable = (target == state.curr_comm) & (state.curr_comm == state.past_comm)
        state.stable_count = torch.where(stable, state.stable_count + 1,
                                         torch.zeros_like(state.stable_count))
        state.active &= state.stable_count < 3
    elif cfg.early_term in (2, 4):
        moved = target != state.curr_comm
        state.move_prob = torch.where(
            moved, torch.ones_like(state.move_prob),
            state.move_prob * (1.0 - cfg.et_delta))
        # draw on the compute device (a per-iteration host round trip here
        # was round-1 weak item 6); rng lives on the same device
        draw = torch.rand(state.move_prob.shape, generator=rng,
                          device=state.move_prob.device)
        state.active = (state.move_prob >= P_CUTOFF) & \
            ((draw < state.move_prob) | moved)
    if cfg.early_term in (3, 4):
        frozen = int((~state.active).sum())
        return int(state.comm.allreduce_scalar(float(frozen)))
    return None


def run_phase(dg: DistGraph, comm: Comm, cfg: LouvainConfig,
              lower: float, threshold: float,
              colors: Optional[torch.Tensor] = None,
              num_colors: int = 0, halo=None):
    """One Louvain phase (local moving to convergence). Returns
    (prev_mod, cvect global labels [nv], iters).
    Ref: distLouvainMethod (louvain.cpp:425-588) and the coloring/ordering
    variants (louvain.cpp:756-2101)."""
    state = PhaseState(dg, comm, halo=halo)
    move_fn = _pick_move_fn(cfg, dg.g.device)
    state.use_hip = (dg.g.device.type == "cuda"
                     and cfg.backend in ("auto", "hip"))
    prev_mod = lower
    iters = 0
    rng = torch.Generator(device=dg.g.device).manual_seed(12345 + comm.rank)
    timers = Timers(sync=dg.g.device.type == "cuda") if cfg.verbose else None

    use_colors = colors is not None and num_colors > 0
    color_order: List[torch.Tensor] = []
    if use_colors:
        for c in range(num_colors):
            color_order.append(torch.nonzero(colors == c, as_tuple=True)[0])

    while iters < cfg.max_iters_per_phase:
        iters += 1
        if timers is None:
            target = _one_sweep(state, cfg, move_fn,
                                color_order if use_colors else None)
            curr_mod = _modularity(state)
        else:
            with timers("sweep"):
                target = _one_sweep(state, cfg, move_fn,
                                    color_order if use_colors else None)
            with timers("modularity"):
                curr_mod = _modularity(state)
            if comm.rank == 0:
                print(f"  iter {iters}: Q={curr_mod:.6f} "
                      f"[{timers.summary()}]")

        frozen = _et_update(state, cfg, target, rng)

        # rotate BEFORE convergence check only in the reference order:
        # check first, rotate after (ref louvain.cpp:541-567)
        if (curr_mod - prev_mod) < threshold:
            break
        prev_mod = max(curr_mod, lower)
        state.past_comm, state.curr_comm = state.curr_comm, target

        if frozen is not None and frozen >= ET_CUTOFF:
            break

    cvect = state.past_comm
    return prev_mod, cvect, iters, state.halo


def _one_sweep(state: PhaseState, cfg: LouvainConfig, move_fn,
               color_order: Optional[List[torch.Tensor]]):
    """One local-moving sweep over all vertices. Plain mode: one simultaneous
    move computation. Coloring mode (-c): sequential passes over color
    classes, with a fresh halo exchange and community update per color
    (ref louvain.cpp:862-901)."""
    nv = state.dg.nv
    dev = state.dg.g.device

    if color_order is None:
        ghost_comm = exchange_ghost_labels(state.halo, state.curr_comm)
        dense, remote_gids, c_size, c_degree, c_gid = state.densify(ghost_comm)
        inp = MoveInputs(state.dg.g.rowptr, state.halo.tails_dense,
                         state.dg.g.weights, dense, state.v_degree,
                         c_size, c_degree, c_gid, state.constant)
        tgt_dense, cw = move_fn(inp)
        target = _dense_to_gid(state, tgt_dense, remote_gids)
        if cfg.early_term:
            target = torch.where(state.active, target, state.curr_comm)
        state.cluster_weight = torch.where(
            state.active, cw, torch.zeros(1, dtype=cw.dtype, device=dev)) \
            if cfg.early_term else cw
        state.apply_moves(target, remote_gids)
        return target

    if cfg.ordering and not cfg.coloring:
        return _ordered_sweep(state, cfg, move_fn, color_order)

    # coloring-ordered (-c): process color classes sequentially, adopting
    # moves after each class, with a fresh ghost-label exchange and remote
    # community-info fetch per class (ref louvain.cpp:862-901).
    work = state.curr_comm.clone()
    cw_total = torch.zeros(nv, dtype=state.dg.g.weights.dtype, device=dev)
    for vidx in color_order:
        ghost_comm = exchange_ghost_labels(state.halo, work)
        saved = state.curr_comm
        state.curr_comm = work
        dense, remote_gids, c_size, c_degree, c_gid = state.densify(ghost_comm)
        state.curr_comm = saved
        inp = MoveInputs(state.dg.g.rowptr, state.halo.tails_dense,
                         state.dg.g.weights, dense, state.v_degree,
                         c_size, c_degree, c_gid, state.constant)
        tgt_dense, cw = move_fn(inp)
        tgt = _dense_to_gid_from(work, tgt_dense, remote_gids,
                                 state.dg.base, state.dg.bound)
        mask = torch.zeros(nv, dtype=torch.bool, device=dev)
        mask[vidx] = True
        if cfg.early_term:
            mask &= state.active
        tgt = torch.where(mask, tgt, work)
        cw_total = torch.where(mask, cw, cw_total)
        # apply deltas for this color class only
        saved_curr = state.curr_comm
        state.curr_comm = work
        state.apply_moves(tgt, remote_gids)
        state.curr_comm = saved_curr
        work = tgt
    state.cluster_weight = cw_total
    return work


def _ordered_sweep(state: PhaseState, cfg: LouvainConfig, move_fn,
                   color_order: List[torch.Tensor]):
    """Vertex-ordering variant (-d, ref distLouvainMethodVertexOrder,
    louvain.cpp:1433-1562): color classes are processed sequentially but
    WITHOUT per-class communication — ghost labels and remote community
    aggregates go stale within the sweep; one exchange at sweep start, one
    delta push at sweep end. All per-class updates happen in the dense
    community id space on device."""
    nv = state.dg.nv
    dev = state.dg.g.device
    ghost_comm = exchange_ghost_labels(state.halo, state.curr_comm)
    dense, remote_gids, c_size, c_degree, c_gid = state.densify(ghost_comm)
    # the per-class updates below mutate these; densify may return live
    # references to state.local_size/local_degree (world-1 fast path)
    c_size = c_size.clone()
    c_degree = c_degree.clone()
    work_dense = dense.clone()
    cw_total = torch.zeros(nv, dtype=state.dg.g.weights.dtype, device=dev)
    for vidx in color_order:
        inp = MoveInputs(state.dg.g.rowptr, state.halo.tails_dense,
                         state.dg.g.weights, work_dense, state.v_degree,
                         c_size, c_degree, c_gid, state.constant)
        tgt_dense, cw = move_fn(inp)
        mask = torch.zeros(nv, dtype=torch.bool, device=dev)
        mask[vidx] = True
        if cfg.early_term:
            mask &= state.active
        new_local = torch.where(mask, tgt_dense, work_dense[:nv])
        cw_total = torch.where(mask, cw, cw_total)
        # local-view aggregate update in dense space (no communication)
        moved = new_local != work_dense[:nv]
        if bool(moved.any()):
            src = work_dense[:nv][moved].to(torch.int64)
            dst = new_local[moved].to(torch.int64)
            vdeg = state.v_degree[moved]
            ones = torch.ones_like(src)
            c_size.index_add_(0, src, -ones)
            c_size.index_add_(0, dst, ones)
            scatter_add_(c_degree, src, -vdeg)
            scatter_add_(c_degree, dst, vdeg)
        work_dense[:nv] = new_local
    state.cluster_weight = cw_total
    target = _dense_to_gid(state, work_dense[:nv], remote_gids)
    state.apply_moves(target, remote_gids)
    return target


def _dense_to_gid(state: PhaseState, dense: torch.Tensor,
                  remote_gids: torch.Tensor) -> torch.Tensor:
    return _dense_to_gid_from(state.curr_comm, dense, remote_gids,
                              state.dg.base, state.dg.bound)


def _dense_to_gid_from(curr: torch.Tensor, dense: torch.Tensor,
                       remote_gids: torch.Tensor, base: int, bound: int):
    nv = curr.numel()
    d = dense.to(torch.int64)
    out = torch.where(d < nv, d + base,
                      remote_gids[(d - nv).clamp(min=0, max=max(remote_gids.numel() - 1, 0))]
                      if remote_gids.numel() else d + base)
    return out


def _modularity(state: PhaseState) -> float:
    if getattr(state, "use_hip", False):
        from . import ops
        parts = ops.modularity_parts(state.cluster_weight, state.local_degree)
    else:
        parts = modularity_parts(state.cluster_weight, state.local_degree)
    state.comm.allreduce_sum_(parts)
    c = state.constant
    return float(parts[0]) * c - float(parts[1]) * c * c


def louvain(dg: DistGraph, comm: Optional[Comm] = None,
            cfg: Optional[LouvainConfig] = None,
            halo=None) -> LouvainResult:
    """Full multi-phase Louvain (ref main.cpp:218-495). Returns the final
    community id per ORIGINAL local vertex (contiguous global ids) and the
    final modularity. `halo`: optional prebuilt HaloContext for the INPUT
    graph (phase 0) — callers that already built one (bench.py's timed
    region) pass it to avoid a duplicate ghost structure (17 GB at s27)."""
    from .coarsen import coarsen, remap_labels
    from .coloring import distance1_coloring

    comm = comm or Comm(dg.g.device)
    cfg = cfg or LouvainConfig()
    dev = dg.g.device

    t_start = time.perf_counter()
    orig_assign = torch.arange(dg.base, dg.bound, device=dev)  # current label of my originals
    prev_mod, curr_mod = -1.0, -1.0
    phase = 0
    short_phase = 0
    tot_iters = 0
    mods = []
    levels: list = []
    level = dg
    times = {"coloring": 0.0, "clustering": 0.0, "rebuild": 0.0}

    while True:
        threshold = _threshold_for_phase(cfg, short_phase) \
            if cfg.threshold_scaling else cfg.threshold

        colors = None
        num_colors = 0
        pre_halo = halo if phase == 0 else None
        if (cfg.coloring or cfg.ordering) and phase == 0:
            from .halo import build_halo as _bh
            t0 = time.perf_counter()
            if pre_halo is None:
                pre_halo = _bh(level, comm)  # shared with PhaseState below
            colors, num_colors = distance1_coloring(
                level, comm, n_hash=max(1, cfg.max_colors // 2),
                halo=pre_halo)
            times["coloring"] += time.perf_counter() - t0

        ne_global = comm.allreduce_scalar(float(level.ne)) \
            if comm.world > 1 else float(level.ne)
        t0 = time.perf_counter()
        halo = None  # phase-0 halo ownership moves to run_phase/coarsen
        curr_mod, cvect, iters, phase_halo = run_phase(
            level, comm, cfg, curr_mod, threshold,
            colors=colors, num_colors=num_colors, halo=pre_halo)
        pre_halo = None
        t_cluster = time.perf_counter() - t0
        times["clustering"] += t_cluster
        tot_iters += iters
        level_entry = {"ne_global": ne_global, "iters": iters,
                       "modularity": curr_mod, "cluster_s": t_cluster,
                       "rebuild_s": 0.0}
        levels.append(level_entry)

        if (curr_mod - prev_mod) > threshold:
            # compose: originals currently assigned to level vertices; level
            # vertex v now belongs to community cvect[v]
            orig_assign = remap_labels(level, comm, orig_assign, cvect)
            mods.append(curr_mod)
            if cfg.one_phase:
                break
            t0 = time.perf_counter()
            if dev.type == "cuda":
                from . import ops
                ops.clear_phase_caches()  # free phase-keyed GPU tensors
            level, renum = coarsen(level, comm, cvect, halo=phase_halo)
            phase_halo = None  # free the ghost structure before next phase
            # cvect-composed orig_assign holds OLD comm gids; renumber them
            orig_assign = renum(orig_assign)
            level_entry["rebuild_s"] = time.perf_counter() - t0
            times["rebuild"] += level_entry["rebuild_s"]
        else:
            if cfg.threshold_scaling and not cfg.one_phase and phase < 10:
                curr_mod2, cvect, iters, _ = run_phase(level, comm, cfg,
                                                       curr_mod, 1.0e-6)
                tot_iters += iters
                level_entry["iters"] += iters
                if (curr_mod2 - curr_mod) > 1.0e-6:
                    orig_assign = remap_labels(level, comm, orig_assign, cvect)
                    mods.append(curr_mod2)
                    curr_mod = curr_mod2
            break

        prev_mod = curr_mod
        curr_mod = -1.0
        phase += 1
        if cfg.threshold_scaling:
            short_phase += 1
        if phase >= cfg.max_phases or tot_iters > MAX_TOTAL_ITERS:
            break

    times["total"] = time.perf_counter() - t_start
    final_mod = mods[-1] if mods else max(prev_mod, curr_mod)
    return LouvainResult(final_mod, orig_assign, phase + 1, tot_iters,
                         times, mods, levels)
