"""Stacked HIP execution engine.

One rank's L node replicas train as a single [L, n] flat parameter stack;
every hot operation — neighbor mixing, dual ascent, fwd/bwd of all L
replicas, fused optimizer steps — is one CDNA4 kernel launch batched over
nodes (ops/hip/*.hip). The reference executes the same math as Python
loops over nodes / parameter tensors / autograd (optimizers/dinno.py:119-
125, dsgd.py:37-58, dsgt.py:58-105); here a full DiNNO round on 8 MNIST
nodes is ~30 launches regardless of node count.

Memory/overhead design points (profile-driven, see profiles/):
* node datasets are resident in HBM; batch assembly is ONE gather kernel
  reading a strided window of a pre-shuffled per-node index stream — no
  per-iteration torch indexing chains;
* neighbor tables are never materialized: kernels take (local stack,
  packed remote buffer) pointer pairs, irecv lands remote rows directly
  in the packed buffer, and the local stack is read in place (no clone,
  no cat in the round path);
* mixing writes into a second stack buffer and the driver pointer-swaps.

Epoch semantics: fixed-size batches from per-node shuffled permutation
streams, wrapping into a fresh permutation at epoch end (the reference's
DataLoader emits one short final batch per epoch instead — documented
deviation, stream-equivalent otherwise).
"""

from __future__ import annotations

import os
import time
from collections import defaultdict

import torch

# NDTA_TIMING=1 prints a host-side wall-time breakdown per section when
# a driver finishes (bench/diagnostic aid; no cost when off)
_TIMING = os.environ.get("NDTA_TIMING", "0") == "1"
_tacc = defaultdict(float)
_tcnt = defaultdict(int)


class _timer:
    __slots__ = ("key", "t0")

    def __init__(self, key):
        self.key = key

    def __enter__(self):
        if _TIMING:
            self.t0 = time.perf_counter()

    def __exit__(self, *a):
        if _TIMING:
            _tacc[self.key] += time.perf_counter() - self.t0
            _tcnt[self.key] += 1


def timing_report():
    if _TIMING:
        tot = sum(_tacc.values())
        print(f"[ndta timing] total {tot*1e3:.1f} ms")
        for k in sorted(_tacc, key=lambda k: -_tacc[k]):
            print(
                f"  {k:20s} {_tacc[k]*1e3:9.1f} ms  {_tcnt[k]:6d} calls"
            )


def set_timing(on: bool):
    """Enable/disable the section timers at runtime (bench.py turns
    them on for its sustained window only, keeping the exactly-timed
    headline region instrumentation-free)."""
    global _TIMING
    _TIMING = bool(on)


def timing_reset():
    _tacc.clear()
    _tcnt.clear()


def timing_dict():
    """Accumulated section times as a JSON-ready dict (ms, calls)."""
    return {
        k: {"ms": round(_tacc[k] * 1e3, 3), "calls": _tcnt[k]}
        for k in sorted(_tacc, key=lambda k: -_tacc[k])
    }

from ..models.spec import ModelSpec, model_spec
from . import get_ext

ACT_IDS = {
    "none": 0,
    "relu": 1,
    "sin_relu": 2,
    "sigmoid": 3,
    "tanh": 4,
    "logsoftmax": 5,
}


def _dataset_tensors(ds):
    """(inputs [N, ...], targets [N]) for TensorDataset/Subset trees."""
    if isinstance(ds, torch.utils.data.TensorDataset):
        return ds.tensors[0], ds.tensors[1]
    if isinstance(ds, torch.utils.data.Subset):
        x, y = _dataset_tensors(ds.dataset)
        idx = torch.as_tensor(ds.indices)
        return x[idx], y[idx]
    if hasattr(ds, "tds"):  # lidar datasets wrap a TensorDataset
        return _dataset_tensors(ds.tds)
    xs, ys = [], []
    for x, y in torch.utils.data.DataLoader(ds, batch_size=1024):
        xs.append(x)
        ys.append(y)
    return torch.cat(xs), torch.cat(ys)


class _StreamSampler:
    """Per-node shuffled index streams; batches are strided views.

    Each node's stream concatenates whole permutations of its dataset;
    a batch is stream[:, pos:pos+B] — zero kernels per draw. Streams are
    rebuilt (host-side, amortized over stream_batches draws) when
    exhausted. Epoch counters advance by consumed/len.
    """

    def __init__(self, lengths, batch, device, seed, epoch_cb,
                 stream_batches=1024):
        self.lengths = lengths
        self.B = batch
        self.S = stream_batches * batch
        self.device = device
        self.epoch_cb = epoch_cb
        # CPU generators fill a pinned staging buffer; ONE async H2D per
        # refill. (Device-side randperms were tried and regressed: each
        # is a multi-launch rocprim sort, ~1.5k launches per refill.)
        self.gens = [
            torch.Generator().manual_seed(seed * 100003 + i)
            for i in range(len(lengths))
        ]
        self.consumed = [0] * len(lengths)
        self.reported_epochs = [0] * len(lengths)
        self.stream = torch.empty(
            len(lengths), self.S, dtype=torch.long, device=device
        )
        self._staging = torch.empty(
            len(lengths), self.S, dtype=torch.long,
            pin_memory=(device.type == "cuda"),
        )
        self.pos = self.S  # force initial fill

    def _refill(self):
        st = self._staging
        for li, n in enumerate(self.lengths):
            filled = 0
            while filled < self.S:
                take = min(n, self.S - filled)
                perm = torch.randperm(n, generator=self.gens[li])
                st[li, filled : filled + take] = perm[:take]
                filled += take
        self.stream.copy_(st, non_blocking=True)
        self.pos = 0

    def next_offset(self) -> int:
        """Advance one batch; refill + epoch bookkeeping; returns the
        stream column offset of the batch."""
        if self.pos + self.B > self.S:
            self._refill()
        off = self.pos
        self.pos += self.B
        for li, n in enumerate(self.lengths):
            self.consumed[li] += self.B
            ep = self.consumed[li] // n
            if self.reported_epochs[li] < ep:
                for _ in range(ep - self.reported_epochs[li]):
                    self.epoch_cb(li)
                self.reported_epochs[li] = ep
        return off

    def next_ref(self):
        """(stream tensor, row stride, offset) for the next batch —
        no per-draw tensor slicing."""
        off = self.next_offset()
        return self.stream, self.S, off


class _OnlineWindowSampler:
    """Sliding-window sampler mirroring OnlineTrajectoryLidarDataset:
    fixed batches from the shuffled current window; advances the window
    (and the dataset's curr_pos, which drives the dynamic graph) when
    the window empties."""

    def __init__(self, datasets, batch, device, seed, epoch_cb):
        self.dss = datasets
        self.B = batch
        self.device = device
        self.epoch_cb = epoch_cb
        self.seed = seed
        # On GPU the per-window shuffle is a keyed Feistel bijection in
        # ONE kernel (ext.feistel_perm) — a device torch.randperm is a
        # rocprim radix-sort + merge chain that cost ~245 us/round
        # amortized (profiles/dens_final_topk.txt). Keyed by
        # (seed, node, refill#): identical streams no matter which rank
        # hosts the node. CPU keeps generator randperm.
        self._use_feistel = device.type == "cuda"
        if not self._use_feistel:
            self.gens = [
                torch.Generator(device=device).manual_seed(
                    seed * 100003 + i
                )
                for i in range(len(datasets))
            ]
        self.refills = [0] * len(datasets)
        # ONE [L, cap] pool tensor: when every node's window has the
        # same size (the standard config — same lidar, same
        # num_scans_in_window) the positions stay synchronized and
        # next_ref is ZERO copies and zero kernels: the gather kernels
        # index the pool directly via (stride=cap, off=pos). Host-side
        # per-node op chains here were 83% of the density round
        # (BENCH r2a timing_breakdown: next_batch 1.83 of 2.18 ms).
        cap = max(
            ds.window_bounds[1] - ds.window_bounds[0] for ds in datasets
        )
        self.cap = cap
        self.pools = torch.empty(
            len(datasets), cap, dtype=torch.long, device=device
        )
        self.wins = [0] * len(datasets)
        self.pos = [0] * len(datasets)
        self.buf = torch.empty(
            len(datasets), batch, dtype=torch.long, device=device
        )
        for li in range(len(datasets)):
            self._refill(li, first=True)

    def _refill(self, li, first=False):
        ds = self.dss[li]
        if not first:
            ds._advance_window()
            if ds.curr_scan_idx <= ds.num_scans_in_window:
                self.epoch_cb(li)  # wrapped around the trajectory
        # device-side shuffle of the window's [lb, ub) index range — the
        # golden path's host list shuffle was ~30 ms per 50k-sample
        # window and dominated the round (profiles/ density pass 2)
        lb, ub = ds.window_bounds
        win = ub - lb
        if win > self.cap:  # dynamic window growth: re-alloc the pool
            self.cap = win
            self.pools = torch.empty(
                len(self.dss), win, dtype=torch.long, device=self.device
            )
            for lj in range(len(self.dss)):
                if lj != li and self.wins[lj] > 0:
                    self._regen(lj)
        self.wins[li] = win
        self._regen(li)
        self.refills[li] += 1
        self.pos[li] = 0

    def _regen(self, li):
        win = self.wins[li]
        lb = self.dss[li].window_bounds[0]
        row = self.pools[li]
        if self._use_feistel:
            key = (self.seed * 100003 + li) * 2654435761 + \
                self.refills[li]
            get_ext().feistel_perm(row, win, lb,
                                   key & 0x7FFFFFFFFFFFFFFF)
        else:
            gen = self.gens[li]
            row[:win] = torch.randperm(
                win, generator=gen, device=self.device
            ) + lb

    def next_ref(self):
        B = self.B
        # fast path: synchronized equal windows, batch fits — no copies
        p0, w0 = self.pos[0], self.wins[0]
        if (p0 + B <= w0
                and all(p == p0 for p in self.pos)
                and all(w == w0 for w in self.wins)):
            for li in range(len(self.dss)):
                self.pos[li] = p0 + B
            return self.pools, self.cap, p0
        for li in range(len(self.dss)):
            p = self.pos[li]
            win = self.wins[li]
            if p + B <= win:
                self.buf[li] = self.pools[li, p : p + B]
                self.pos[li] = p + B
            else:
                head = self.pools[li, p:win].clone()
                self._refill(li)
                take = B - head.numel()
                self.buf[li] = torch.cat(
                    [head, self.pools[li, :take]]
                )
                self.pos[li] = take
        return self.buf, self.B, 0


class StackedEngine:
    def __init__(self, problem):
        self.ext = get_ext()
        self.pr = problem
        self.device = problem.device
        self.dtype = torch.get_default_dtype()
        if self.dtype not in (torch.float32, torch.float64):
            raise ValueError(
                "stacked engine supports fp32/fp64 (precision knob)"
            )

        self.local_nodes = problem.local_nodes
        self.L = len(self.local_nodes)
        self.n = problem.n
        first = self.local_nodes[0]
        self.spec: ModelSpec = model_spec(problem.models[first])

        rows = [
            torch.nn.utils.parameters_to_vector(
                problem.models[i].parameters()
            )
            .detach()
            .to(self.device, self.dtype)
            for i in self.local_nodes
        ]
        self.theta = torch.stack(rows).contiguous()
        self.grad = torch.zeros_like(self.theta)

        self._stage_data()
        self.B = problem.conf["train_batch_size"]
        self._has_node0 = 0 in self.local_nodes
        self._zero_plan = None
        self._bufs = None
        self._loss_kind = type(problem.base_loss).__name__  # NLLLoss etc.
        # True while self.grad is all zeros (init, or after a step
        # kernel that cleared it); lets fwd_bwd skip the fill launch
        self.grad_is_zero = True
        # spec tables for the C++ fwd/bwd chains (ONE pybind call per
        # pass instead of one per layer-op: ~10 us of host overhead per
        # ext call made the host the MNIST round bottleneck — see
        # ext.hip fwd_chain). NDTA_PY_CHAIN=1 selects the python
        # per-op loops (debug / A-B reference).
        self._use_chain = os.environ.get("NDTA_PY_CHAIN", "0") != "1"
        self._spec_t = None
        self._scales_t = None
        # device-resident train-loss EMA (the reference's tloss tracker,
        # problems/dist_online_dense_problem.py:129-137, without a
        # host sync per round)
        self.tloss_dev = None

    # ------------------------------------------------------------------
    def update_tloss(self, loss_l: torch.Tensor):
        d = float(self.pr.tloss_decay)
        if self.tloss_dev is None:
            self.tloss_dev = loss_l.clone()
        else:
            self.tloss_dev = torch.where(
                self.tloss_dev == 0,
                loss_l,
                (1 - d) * self.tloss_dev + d * loss_l,
            )

    # ------------------------------------------------------------------
    def _stage_data(self):
        pr = self.pr
        xs, ys = [], []
        for i in self.local_nodes:
            x, y = _dataset_tensors(pr.train_sets[i])
            xs.append(x.reshape(x.shape[0], -1).to(self.device, self.dtype))
            ys.append(y.to(self.device))
        self.lengths = [x.shape[0] for x in xs]
        maxlen = max(self.lengths)
        feat = xs[0].shape[1]
        self.X_all = torch.zeros(
            self.L, maxlen, feat, device=self.device, dtype=self.dtype
        )
        is_class = ys[0].dtype in (torch.int64, torch.int32)
        self.Y_all = torch.zeros(
            self.L,
            maxlen,
            device=self.device,
            dtype=torch.long if is_class else self.dtype,
        )
        for li, (x, y) in enumerate(zip(xs, ys)):
            self.X_all[li, : x.shape[0]] = x
            self.Y_all[li, : y.shape[0]] = (
                y.long() if is_class else y.to(self.dtype)
            )
        self.classification = is_class

        def epoch_cb(li):
            pr.epoch_tracker[self.local_nodes[li]] += 1

        seed = int(pr.conf.get("data_seed", 0))
        B = pr.conf["train_batch_size"]
        online = [
            pr.train_sets[i]
            for i in self.local_nodes
            if hasattr(pr.train_sets[i], "curr_idx_list")
        ]
        if len(online) == self.L and self.L > 0:
            self.sampler = _OnlineWindowSampler(
                online, B, self.device, seed, epoch_cb
            )
        else:
            self.sampler = _StreamSampler(
                self.lengths, B, self.device, seed, epoch_cb
            )

    # ------------------------------------------------------------------
    def _alloc_bufs(self, M=None, train=True):
        """Workspace pack for per-node batch size M (cached by M)."""
        L = self.L
        M = M if M is not None else self.B
        mk = lambda el: torch.empty(
            L * M, el, device=self.device, dtype=self.dtype
        )
        acts, zs, dzs, idxs = [], [], [], []
        for layer in self.spec.layers:
            acts.append(mk(layer.out_elems))
            # sin_relu needs z in backward, but for a tiny in_dim the
            # dx epilogue RECOMPUTES it (4 flops) instead of storing /
            # re-reading an [L*M, out] tensor (kernel z-recompute mode)
            zs.append(
                mk(layer.out_elems)
                if (train and layer.activation == "sin_relu"
                    and layer.in_dim > 4)
                else None
            )
            dzs.append(mk(layer.out_elems) if train else None)
            idxs.append(
                torch.empty(
                    L * M, layer.out_elems, device=self.device,
                    dtype=torch.uint8,
                )
                if layer.kind == "conv_pool"
                else None
            )
        pack = {
            "M": M,
            "acts": acts, "zs": zs, "dzs": dzs, "idxs": idxs,
            "xb": mk(self.spec.in_elems),
            "yb": torch.empty(
                L * M, device=self.device,
                dtype=torch.long if self.classification else self.dtype,
            ),
            "logp": mk(self.spec.layers[-1].out_elems)
            if self.classification else None,
            "loss": torch.zeros(L, device=self.device, dtype=self.dtype),
        }
        return pack

    # ------------------------------------------------------------------
    # fully fused MNIST train step (ops/hip/fused_mnist.hip): one launch
    # per primal iteration instead of the ~12-kernel layered chain.
    # DEFAULT OFF — measured SLOWER than the layered path on MI355X
    # (best 2263 vs 3650 rounds/s at the 8-node bench): per-tile fusion
    # re-streams fc1's 221KB weight panel per block (cost grows as tiles
    # shrink) while large tiles leave 3/4 of the 256-CU chip idle at
    # 1 block/CU. The layered GEMMs share weight panels across the whole
    # batch and fill the chip. Kept as a validated alternative
    # (NDTA_FUSED=1; numerics-tested against the layered path).
    def fused_step_available(self) -> bool:
        if os.environ.get("NDTA_FUSED", "0") != "1":
            return False
        if self.spec.name != "mnist_conv" or not self.classification:
            return False
        if not isinstance(self.sampler, _StreamSampler):
            return False
        return self.device.type == "cuda"

    def run_fused_mnist(self, off=0, graph_offs=None, pit=0,
                        want_loss=False):
        """One-launch train step writing per-tile gradient SLABS
        (plain stores — no atomics, no pre-zero); the fused optimizer
        step reduces the slabs on the fly (nparts), other consumers
        call reduce_fused_grad()."""
        conv, fc1, fc2 = self.spec.layers
        ti = int(os.environ.get("NDTA_FUSED_TI", "8"))
        nt = (self.B + ti - 1) // ti
        if (getattr(self, "grad_parts", None) is None
                or self.grad_parts.shape[1] != nt):
            self.grad_parts = torch.empty(
                self.L, nt, self.n, device=self.device,
                dtype=self.dtype,
            )
        self.fused_nparts = nt
        loss_buf = None
        if want_loss:
            if self._bufs is None:
                self._bufs = self._alloc_bufs()
            self._bufs["loss"].zero_()
            loss_buf = self._bufs["loss"]
        self.ext.mnist_train_step(
            self.X_all, self.Y_all, self.sampler.stream, graph_offs,
            self.theta, self.grad_parts, loss_buf, pit, off,
            self.sampler.S,
            conv.w_off, conv.b_off, fc1.w_off, fc1.b_off, fc2.w_off,
            fc2.b_off, self.B, conv.out_dim, conv.kernel_size,
            conv.in_dim, fc1.out_dim, fc2.out_dim, ti, 1.0,
        )
        return loss_buf

    def reduce_fused_grad(self):
        self.ext.reduce_parts(
            self.grad_parts, self.grad, self.fused_nparts
        )

    def fused_advance(self):
        """Advance the sampler for a fused step; returns the offset."""
        off = self.sampler.next_offset()
        if self._has_node0:
            self.pr.forward_cnt += self.B
        return off

    # ------------------------------------------------------------------
    def next_batch(self):
        """Assemble the next per-node batches with two gather kernels.
        On the fused-fc path the conv kernels gather the image rows
        themselves (conv_pool_*_idx): no gather launch, no xb buffer.
        """
        if self._bufs is None:
            self._bufs = self._alloc_bufs()
        idx, stride, off = self.sampler.next_ref()
        self._last_off = off
        if self._has_node0:
            self.pr.forward_cnt += self.B
        if self.fc_block_applicable():
            return None, None
        xb, yb = self._bufs["xb"], self._bufs["yb"]
        self.ext.gather_batch(self.X_all, idx, xb, stride, off)
        if self.classification:
            # targets stay resident: the fused NLL kernel gathers them
            yb = None
        else:
            self.ext.gather_targets(self.Y_all, idx, yb, stride, off)
        return xb, yb

    # ------------------------------------------------------------------
    def _chain_meta(self):
        if self._spec_t is None:
            rows, scales = [], []
            for layer in self.spec.layers:
                rows.append([
                    1 if layer.kind == "conv_pool" else 0,
                    layer.w_off, layer.b_off, layer.in_dim,
                    layer.out_dim, ACT_IDS[layer.activation],
                    getattr(layer, "kernel_size", 0) or 0,
                ])
                scales.append(float(getattr(layer, "scale", 1.0)))
            self._spec_t = torch.tensor(rows, dtype=torch.long)
            self._scales_t = torch.tensor(scales, dtype=torch.float64)
        return self._spec_t, self._scales_t

    def forward(self, xb, bufs=None, train_skip_logp=False):
        if self._bufs is None:
            self._bufs = self._alloc_bufs()
        bufs = bufs if bufs is not None else self._bufs
        ext = self.ext
        cur = xb
        M = bufs["M"]
        if self._use_chain:
            spec_t, scales_t = self._chain_meta()
            ext.fwd_chain(
                spec_t, scales_t, None, None, 0, 0, xb, self.theta,
                bufs["acts"], bufs["zs"], bufs["idxs"], bufs["logp"],
                M, train_skip_logp,
            )
            last = self.spec.layers[-1]
            if last.activation == "logsoftmax" and not train_skip_logp:
                return bufs["logp"]
            return bufs["acts"][-1]
        for li, layer in enumerate(self.spec.layers):
            out = bufs["acts"][li]
            if layer.kind == "conv_pool":
                ext.conv_pool_fwd(
                    cur, self.theta, out, bufs["idxs"][li],
                    layer.w_off, layer.b_off, M, layer.out_dim,
                    layer.kernel_size, layer.in_dim,
                )
            else:
                act = layer.activation
                if act == "logsoftmax":
                    ext.linear_fwd(
                        cur, self.theta, out, None, layer.w_off,
                        layer.b_off, M, layer.in_dim, layer.out_dim,
                        ACT_IDS["none"], 1.0,
                    )
                    # training backward consumes the LOGITS through the
                    # fused NLL kernel; log-probs are only materialized
                    # for evaluation
                    if not train_skip_logp:
                        ext.logsoftmax(out, bufs["logp"], layer.out_dim)
                else:
                    ext.linear_fwd(
                        cur, self.theta, out, bufs["zs"][li],
                        layer.w_off, layer.b_off, M, layer.in_dim,
                        layer.out_dim, ACT_IDS[act], layer.scale,
                    )
            # under train_skip_logp the logp buffer was never written —
            # hand back the logits (the training path consumes those
            # through the fused NLL kernel anyway)
            cur = (
                out
                if layer.activation != "logsoftmax" or train_skip_logp
                else bufs["logp"]
            )
        return cur

    # ------------------------------------------------------------------
    def fc_block_applicable(self):
        """One-launch fused fc block (fused_mnist.hip fc_block_k):
        conv -> [fc1+fc2 fwd + NLL + full fc bwd] -> conv bwd.
        Replaces 7 floor-bound small kernels per primal iteration
        (~65 us -> ~15 us of GPU time at the MNIST bench shapes)."""
        if os.environ.get("NDTA_FC_BLOCK", "1") == "0":
            return False
        ls = self.spec.layers
        return (
            self.device.type == "cuda"
            and self.classification
            and len(ls) == 3
            and ls[0].kind == "conv_pool"
            # dX0 carries the conv block's relu' mask inside fc_block_k
            and getattr(ls[0], "activation", "relu") == "relu"
            and ls[1].kind == "linear"
            and ls[2].kind == "linear"
            and ls[1].activation == "relu"
            and ls[2].activation == "logsoftmax"
            and ls[1].out_dim <= 64
            and ls[2].out_dim <= 16
        )

    def fwd_bwd(self, xb, yb, loss_scale=1.0, want_loss=False,
                graph_offs=None, pit=0):
        """Forward + backward for one primal iteration; takes the fused
        fc-block path when the model/loss shape allows it."""
        if graph_offs is None and self.fc_block_applicable():
            bufs = self._bufs
            conv, fc1, fc2 = self.spec.layers
            M = self.B
            ext = self.ext
            idx_t = self.sampler.stream if hasattr(
                self.sampler, "stream") else self.sampler.buf
            stride = self.sampler.S if hasattr(
                self.sampler, "S") else self.B
            ext.conv_pool_fwd_idx(
                self.X_all, idx_t, stride, self._last_off,
                self.theta, bufs["acts"][0], bufs["idxs"][0],
                conv.w_off, conv.b_off, M, conv.out_dim,
                conv.kernel_size, conv.in_dim,
            )
            # fc grads accumulate atomically: the stack must be all
            # zeros here. The step kernels clear it as they consume it
            # (zero_grad), so the fill launch is usually skipped.
            if not self.grad_is_zero:
                self.grad.zero_()
            self.grad_is_zero = False
            loss_buf = None
            if want_loss:
                bufs["loss"].zero_()
                loss_buf = bufs["loss"]
            ext.fc_block(
                bufs["acts"][0], self.theta, self.Y_all, idx_t,
                stride, self._last_off, self.grad, bufs["dzs"][0],
                bufs["dzs"][1],
                loss_buf, fc1.w_off, fc1.b_off, fc2.w_off, fc2.b_off,
                M, fc1.in_dim, fc1.out_dim, fc2.out_dim, loss_scale,
                True,
            )
            ext.conv_pool_bwd_idx(
                bufs["dzs"][0], bufs["idxs"][0], self.X_all, idx_t,
                stride, self._last_off, self.grad,
                conv.w_off, conv.b_off, M, conv.out_dim,
                conv.kernel_size, conv.in_dim,
            )
            return loss_buf if want_loss else None
        self.forward(xb, train_skip_logp=True)
        return self.backward(
            xb, yb, loss_scale=loss_scale, want_loss=want_loss,
            graph_offs=graph_offs, pit=pit,
        )

    # ------------------------------------------------------------------
    def backward(self, xb, yb, loss_scale=1.0, want_loss=False,
                 graph_offs=None, pit=0):
        bufs = self._bufs
        ext = self.ext
        layers = self.spec.layers
        nl = len(layers)
        M = self.B
        # atomic-accumulating layers need their grad slices zeroed;
        # store-path layers overwrite. Zero the whole stack only when a
        # big atomic linear layer exists (density), else just the conv
        # slices (MNIST: 78 of 28440 elements).
        if self._zero_plan is None:
            slices = []
            whole = False
            for layer in self.spec.layers:
                if layer.kind == "conv_pool":
                    cnt = (layer.out_dim * layer.kernel_size**2
                           + layer.out_dim)
                    slices.append((layer.w_off, cnt))
                elif (M >= 256 and layer.in_dim >= 16
                      and layer.out_dim >= 16) or M > 2048:
                    whole = True
            self._zero_plan = ("whole",) if whole else ("slices", slices)
        if self._use_chain:
            zero_mode = 1 if self._zero_plan[0] == "whole" else 2
            loss_buf = bufs["loss"] if want_loss else None
            last = layers[-1]
            if self.classification:
                idx_t = (
                    self.sampler.stream
                    if hasattr(self.sampler, "stream")
                    else self.sampler.buf
                )
                stride = (
                    self.sampler.S
                    if hasattr(self.sampler, "S") else self.B
                )
                loss_kind = 0
                yb_arg = None
                idx_off = 0 if graph_offs is not None else self._last_off
            else:
                idx_t, stride, idx_off = None, 0, 0
                if (self._loss_kind == "BCELoss"
                        and last.activation == "sigmoid"):
                    loss_kind = 1
                elif self._loss_kind == "MSELoss":
                    loss_kind = 2
                else:
                    loss_kind = 3
                yb_arg = yb
            spec_t, scales_t = self._chain_meta()
            ext.bwd_chain(
                spec_t, scales_t, xb, self.theta, self.grad,
                bufs["acts"], bufs["zs"], bufs["dzs"], bufs["idxs"],
                loss_kind, self.Y_all if self.classification else None,
                idx_t, graph_offs, pit, stride, idx_off, yb_arg,
                loss_buf, loss_scale, M, zero_mode,
            )
            return loss_buf if want_loss else None

        if self._zero_plan[0] == "whole":
            self.grad.zero_()
        else:
            for off, cnt in self._zero_plan[1]:
                self.grad[:, off : off + cnt].zero_()
        loss_buf = None
        if want_loss:
            bufs["loss"].zero_()
            loss_buf = bufs["loss"]

        last = layers[-1]
        dz = bufs["dzs"][nl - 1]
        if self.classification:
            # fused: logits -> row LSE -> resident-target gather -> dZ
            idx_t = (
                self.sampler.stream
                if hasattr(self.sampler, "stream")
                else self.sampler.buf
            )
            stride = (
                self.sampler.S if hasattr(self.sampler, "S") else self.B
            )
            ext.nll_fused(
                bufs["acts"][nl - 1], self.Y_all, idx_t, dz, loss_buf,
                graph_offs, pit, stride,
                0 if graph_offs is not None else self._last_off,
                last.out_dim, M, loss_scale,
            )
        else:
            kind = self._loss_kind
            yhat = bufs["acts"][nl - 1].reshape(-1)
            if kind == "BCELoss" and last.activation == "sigmoid":
                ext.bce_bwd(yhat, yb, dz.reshape(-1), loss_buf, M,
                            loss_scale)
            else:
                mode = 0 if kind == "MSELoss" else 1
                dy = torch.empty_like(dz)
                ext.regression_bwd(
                    yhat, yb, dy.reshape(-1), loss_buf, M, loss_scale,
                    mode,
                )
                ext.act_grad(
                    dy, bufs["acts"][nl - 1], None, dz,
                    ACT_IDS[last.activation], last.scale,
                )

        for li in range(nl - 1, -1, -1):
            layer = layers[li]
            below = bufs["acts"][li - 1] if li > 0 else xb
            dz = bufs["dzs"][li]
            if layer.kind == "conv_pool":
                ext.conv_pool_bwd(
                    dz, bufs["idxs"][li], below, self.grad,
                    layer.w_off, layer.b_off, M, layer.out_dim,
                    layer.kernel_size, layer.in_dim,
                )
                continue  # conv is the first layer: no dX
            ext.linear_bwd_dw(
                dz, below, self.grad, layer.w_off, layer.b_off, M,
                layer.in_dim, layer.out_dim,
            )
            if li > 0:
                # dX with the below layer's activation bwd fused into
                # the epilogue (no separate act_grad pass). For a
                # sin_relu below layer with tiny in_dim (FourierNet
                # encode), z_below is RECOMPUTED from the below input
                # in the epilogue — no [M, I] Zb tensor is stored in
                # forward or read back here (2x 328 MB/round saved in
                # the density config).
                dz_below = bufs["dzs"][li - 1]
                lb = layers[li - 1]
                act_b = (
                    ACT_IDS[lb.activation]
                    if lb.activation not in ("none", "logsoftmax")
                    else 0
                )
                xb2, wb_off, bb_off, ib = None, 0, 0, 0
                if (act_b == ACT_IDS["sin_relu"]
                        and bufs["zs"][li - 1] is None):
                    xb2 = bufs["acts"][li - 2] if li - 1 > 0 else xb
                    wb_off, bb_off = lb.w_off, lb.b_off
                    ib = lb.in_dim
                ext.linear_bwd_dx(
                    dz, self.theta, dz_below,
                    bufs["acts"][li - 1] if act_b else None,
                    bufs["zs"][li - 1], act_b, lb.scale,
                    layer.w_off, M, layer.in_dim, layer.out_dim,
                    xb2, wb_off, bb_off, ib,
                )
        return bufs["loss"] if want_loss else None

    # ------------------------------------------------------------------
    # batched validation: all local nodes evaluate the shared val set in
    # one stacked forward per chunk (the eager per-node torch loop cost
    # ~130 ms per evaluation at the paper config — 400x a training round)
    def _stage_val(self):
        if getattr(self, "_val", None) is not None:
            return self._val
        pr = self.pr
        x, y = _dataset_tensors(pr.val_set)
        xd = x.reshape(x.shape[0], -1).to(self.device, self.dtype)
        if self.classification:
            yd = y.long().to(self.device)
        else:
            yd = y.to(self.device, self.dtype)
        self._val = (xd, yd)
        self._val_pack = None
        return self._val

    def validate_all(self):
        """Per-node validation on the shared val set.

        Classification: (loss [L], top1 acc [L], correct [L, V] bool) —
        replicating the reference's metric convention of summing batch
        means then dividing by the dataset size
        (problems/dist_mnist_problem.py:111-132).
        Regression: summed batch-mean losses [L]
        (dist_dense_problem.py:119-133).
        """
        pr = self.pr
        xd, yd = self._stage_val()
        V = xd.shape[0]
        Bv = min(int(pr.conf.get("val_batch_size", 1024)), V)
        if self._val_pack is None or self._val_pack["M"] != Bv:
            self._val_pack = self._alloc_bufs(M=Bv, train=False)
        pack = self._val_pack
        L = self.L
        loss_sum = torch.zeros(L, device=self.device, dtype=self.dtype)
        if self.classification:
            correct = torch.zeros(
                L, V, device=self.device, dtype=torch.bool
            )
        nchunks = 0
        for v0 in range(0, V, Bv):
            v1 = min(V, v0 + Bv)
            c = v1 - v0
            xb = pack["xb"]
            chunk = xd[v0:v1]
            if c < Bv:  # ragged tail: pad with the first row
                chunk = torch.cat(
                    [chunk, xd[:1].expand(Bv - c, -1)], dim=0
                )
            xb.copy_(chunk.repeat(L, 1))
            out = self.forward(xb, bufs=pack)
            nchunks += 1
            if self.classification:
                logp = out.reshape(L, Bv, -1)[:, :c]
                yv = yd[v0:v1]
                loss_sum += torch.nn.functional.nll_loss(
                    logp.reshape(L * c, -1),
                    yv.repeat(L),
                    reduction="none",
                ).reshape(L, c).mean(dim=1)
                correct[:, v0:v1] = logp.argmax(dim=2) == yv
            else:
                yhat = out.reshape(L, Bv)[:, :c]
                yv = yd[v0:v1]
                kind = self._loss_kind
                if kind == "BCELoss":
                    eps = 1e-12
                    pcl = yhat.clamp(eps, 1 - eps)
                    bl = -(
                        yv * pcl.log() + (1 - yv) * (1 - pcl).log()
                    ).mean(dim=1)
                elif kind == "MSELoss":
                    bl = ((yhat - yv) ** 2).mean(dim=1)
                else:
                    bl = (yhat - yv).abs().mean(dim=1)
                loss_sum += bl
        if self.classification:
            acc = correct.sum(dim=1).to(self.dtype) / V
            return loss_sum.cpu() / V, acc.cpu(), correct.cpu()
        return loss_sum.cpu()

    # ------------------------------------------------------------------
    def flush_to_models(self):
        for li, i in enumerate(self.local_nodes):
            torch.nn.utils.vector_to_parameters(
                self.theta[li].to(torch.get_default_dtype()),
                self.pr.models[i].parameters(),
            )

    # ------------------------------------------------------------------
    # neighbor-table plumbing (delegates to parallel/schedule.py so the
    # multi-rank plan is unit-testable on CPU with gloo)
    def remote_plan(self, width_factor=1):
        from ..parallel import schedule

        return schedule.remote_plan(
            self.pr.comm, self.pr.layout, self.pr.graph, self.n,
            self.device, self.dtype, width_factor,
        )

    def build_csr(self, row_of, include_self=False, W=None):
        from ..parallel import schedule

        return schedule.build_csr(
            self.pr.graph, self.local_nodes, row_of, self.device,
            self.dtype, include_self=include_self, W=W,
        )

    def row_map(self, remote_nodes):
        from ..parallel import schedule

        return schedule.row_map(self.local_nodes, remote_nodes)

    def degrees(self) -> torch.Tensor:
        from ..parallel import schedule

        return schedule.degrees(
            self.pr.graph, self.local_nodes, self.device
        )


# ======================================================================
# Optimizer drivers on the stacked engine
# ======================================================================

_OPT_MODE = {"adam": 0, "adamw": 1, "sgd": 2}


def _graph_is_static(pr):
    from ..problems.base import ProblemBase

    return type(pr).update_graph is ProblemBase.update_graph


def _edge_key(pr):
    """Hashable identity of the current communication graph; dynamic
    graphs change edges only when robots cross the comm radius, so
    plans/CSRs are cached on this key instead of rebuilt every round."""
    return tuple(sorted(map(tuple, map(sorted, pr.graph.edges()))))


class DiNNOStackedDriver:
    """DiNNO outer loop with the fused kernels (same math as
    optimizers/dinno.py).

    Isolated nodes (possible under dynamic graphs) are frozen whole —
    theta, Adam moments and duals — matching the golden engine's skip
    (the reference crashes on them). One deliberate corner: with
    ``persistant_primal_opt`` the bias-correction step count is global,
    so a node that reconnects after an isolated stretch sees a larger
    t than the golden engine's per-node Adam would give it; the
    non-persistent mode (the paper configs) matches exactly.
    """

    def __init__(self, dinno, pr):
        self.opt = dinno
        self.pr = pr
        self.eng: StackedEngine = pr.stacked

    def prepare(self):
        eng = self.eng
        conf = self.opt.conf
        self.mode = _OPT_MODE[conf["primal_optimizer"]]
        self.persistent = conf["persistant_primal_opt"]
        self.wd = 0.01 if self.mode == 1 else 0.0
        self.pits = conf["primal_iterations"]
        self.duals = torch.zeros_like(eng.theta)
        self.s = torch.zeros_like(eng.theta)
        self.m = torch.zeros_like(eng.theta)
        self.v = torch.zeros_like(eng.theta)
        self.rho = conf["rho_init"]
        self.step_t = 0
        self._plans = {}
        # hipGraph capture of the whole round (NDTA_GRAPHS=1 enables).
        # Measured on MI355X: NEUTRAL to -4% vs eager at both 8-node and
        # 2-node packings — the eager host launches already hide under
        # GPU time, and the remaining floor is the per-kernel dependent
        # boundary (~1.5 us each), which replay does not remove. Kept as
        # a validated capability (tests/test_stacked_gpu.py exercises
        # capture+replay parity) for deployments where host python is
        # contended.
        self._graph = None
        self._graph_warm = 0
        self._graph_failed = False
        self.graph_mode = (
            eng.device.type == "cuda"
            and os.environ.get("NDTA_GRAPHS", "0") == "1"
            and _graph_is_static(self.pr)
            and isinstance(eng.sampler, _StreamSampler)
            and not bool(getattr(self.pr, "track_tloss", False))
        )
        if self.graph_mode:
            pits = self.pits
            dt = eng.dtype
            self._sched_dev = torch.zeros(
                2 + 2 * pits, device=eng.device, dtype=dt
            )
            self._sched_host = torch.zeros(2 + 2 * pits, dtype=dt)
            self._offs_dev = torch.zeros(
                pits, device=eng.device, dtype=torch.long
            )
            self._offs_host = torch.zeros(pits, dtype=torch.long)
            try:
                self._sched_host = self._sched_host.pin_memory()
                self._offs_host = self._offs_host.pin_memory()
            except RuntimeError:
                pass

    def _round_plan(self):
        key = _edge_key(self.pr)
        plan = self._plans.get(key)
        if plan is not None:
            return plan
        eng = self.eng
        remote_nodes, rbuf, dests = eng.remote_plan()
        row_of = eng.row_map(remote_nodes)
        offs, idx, _ = eng.build_csr(row_of, include_self=False)
        deg = eng.degrees()
        plan = (rbuf, dests, offs, idx, deg)
        if len(self._plans) < 256:
            self._plans[key] = plan
        return plan

    def _graph_body(self, rbuf, offs, idx, deg):
        """One DiNNO round with all round-varying scalars read from
        device buffers — every launch has constant args, so the whole
        sequence is hipGraph-capturable."""
        eng = self.eng
        ext = eng.ext
        bufs = eng._bufs
        ext.dinno_dual_threg_sched(
            eng.theta, rbuf, offs, idx, self.duals, self.s,
            self._sched_dev,
        )
        fused = eng.fused_step_available()
        xb = bufs["xb"]
        for pit in range(self.pits):
            if fused:
                eng.run_fused_mnist(
                    graph_offs=self._offs_dev, pit=pit
                )
                grad_t, nparts = eng.grad_parts, eng.fused_nparts
            else:
                ext.gather_batch_dev(
                    eng.X_all, eng.sampler.stream, xb, self._offs_dev,
                    pit, eng.sampler.S,
                )
                eng.fwd_bwd(
                    xb, None, graph_offs=self._offs_dev, pit=pit
                )
                grad_t, nparts = eng.grad, 1
            first = (not self.persistent) and pit == 0
            ext.fused_step_sched(
                eng.theta, grad_t, self.duals, self.s, deg,
                None if self.mode == 2 else self.m,
                None if self.mode == 2 else self.v,
                self._sched_dev, pit, 0.9, 0.999, 1e-8, self.wd,
                self.mode, first, nparts,
            )

    def _step_round_graph(self, k):
        import math as _math

        opt, pr, eng = self.opt, self.pr, self.eng
        self.rho *= opt.rho_scaling
        pr.update_graph()
        rbuf, dests, offs, idx, deg = self._round_plan()
        if pr.comm.world > 1:
            pr.comm.exchange_rows(
                pr.layout, list(pr.graph.edges()), [eng.theta], [dests]
            )
        if eng._bufs is None:
            eng._bufs = eng._alloc_bufs()

        # host-side schedule for this round
        lr = float(
            opt.primal_lr[0] if self.persistent else opt.primal_lr[k]
        )
        sh = self._sched_host
        sh[0] = self.rho
        sh[1] = lr
        for pit in range(self.pits):
            if self.persistent:
                self.step_t += 1
                t = self.step_t
            else:
                t = pit + 1
            sh[2 + 2 * pit] = 1.0 - _math.pow(0.9, t)
            sh[3 + 2 * pit] = 1.0 - _math.pow(0.999, t)
            self._offs_host[pit] = eng.sampler.next_offset()
        if eng._has_node0:
            pr.forward_cnt += eng.B * self.pits
        self._sched_dev.copy_(self._sched_host, non_blocking=True)
        self._offs_dev.copy_(self._offs_host, non_blocking=True)

        if self._graph is not None:
            self._graph.replay()
            return
        if self._graph_warm < 2:
            # warmup executions on a side stream (these ARE real rounds)
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                self._graph_body(rbuf, offs, idx, deg)
            torch.cuda.current_stream().wait_stream(side)
            self._graph_warm += 1
            return
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._graph_body(rbuf, offs, idx, deg)
            self._graph = g
            self._graph.replay()  # this round's work
        except Exception as e:  # noqa: BLE001
            print(f"[ndta] hipGraph capture failed ({e}); "
                  "falling back to eager rounds", flush=True)
            self.graph_mode = False
            self._graph_failed = True
            self._graph_body(rbuf, offs, idx, deg)

    def step_round(self, k):
        opt, pr, eng = self.opt, self.pr, self.eng
        ext = eng.ext
        if self.graph_mode:
            return self._step_round_graph(k)

        self.rho *= opt.rho_scaling
        with _timer("update_graph"):
            pr.update_graph()
        with _timer("round_plan"):
            rbuf, dests, offs, idx, deg = self._round_plan()
        with _timer("exchange"):
            if pr.comm.world > 1:
                pr.comm.exchange_rows(
                    pr.layout, list(pr.graph.edges()), [eng.theta],
                    [dests],
                )
        with _timer("dual_threg"):
            ext.dinno_dual_threg(
                eng.theta, rbuf, offs, idx, self.duals, self.s, self.rho
            )

        if not self.persistent:
            # the kernel's first_step flag zeroes the moments logically
            self.step_t = 0
        lr = float(
            opt.primal_lr[0] if self.persistent else opt.primal_lr[k]
        )

        want_tl = bool(getattr(pr, "track_tloss", False))
        fused = eng.fused_step_available()
        for pi in range(self.pits):
            wl = want_tl and pi == self.pits - 1
            if fused:
                with _timer("train_step"):
                    off = eng.fused_advance()
                    lb = eng.run_fused_mnist(off=off, want_loss=wl)
                grad_t, nparts = eng.grad_parts, eng.fused_nparts
            else:
                with _timer("next_batch"):
                    xb, yb = eng.next_batch()
                with _timer("fwd_bwd"):
                    lb = eng.fwd_bwd(xb, yb, want_loss=wl)
                grad_t, nparts = eng.grad, 1
            if wl and lb is not None:
                eng.update_tloss(lb)
            with _timer("fused_step"):
                self.step_t += 1
                zg = nparts == 1 and grad_t is eng.grad
                ext.fused_step(
                    eng.theta, grad_t, self.duals, self.s, deg,
                    None if self.mode == 2 else self.m,
                    None if self.mode == 2 else self.v,
                    self.rho, lr, 0.9, 0.999, 1e-8, self.wd,
                    self.step_t, self.mode, self.step_t == 1, nparts,
                    zg,
                )
                if zg:
                    eng.grad_is_zero = True

    def _opt_state(self):
        return {
            "duals": self.duals, "s": self.s, "m": self.m, "v": self.v,
            "rho": self.rho, "step_t": self.step_t,
        }

    def _load_opt_state(self, st):
        self.duals.copy_(st["duals"])
        self.s.copy_(st["s"])
        self.m.copy_(st["m"])
        self.v.copy_(st["v"])
        self.rho = st["rho"]
        self.step_t = st["step_t"]

    def run(self, profiler=None):
        _run_rounds(self, profiler)


class DSGDStackedDriver:
    """DSGD round (synchronous mixing) with fused kernels."""

    def __init__(self, dsgd, pr):
        self.opt = dsgd
        self.pr = pr
        self.eng: StackedEngine = pr.stacked

    def prepare(self):
        self.alph = self.opt.alph0
        self.theta_next = torch.empty_like(self.eng.theta)
        self._plans = {}

    def _round_plan(self):
        from ..utils import graph_generation

        key = _edge_key(self.pr)
        plan = self._plans.get(key)
        if plan is not None:
            return plan
        eng = self.eng
        W = graph_generation.get_metropolis(self.pr.graph)
        remote_nodes, rbuf, dests = eng.remote_plan()
        row_of = eng.row_map(remote_nodes)
        offs, idx, w = eng.build_csr(row_of, include_self=True, W=W)
        plan = (rbuf, dests, offs, idx, w)
        if len(self._plans) < 256:
            self._plans[key] = plan
        return plan

    def step_round(self, k):
        opt, pr, eng = self.opt, self.pr, self.eng
        ext = eng.ext
        pr.update_graph()
        rbuf, dests, offs, idx, w = self._round_plan()
        self.alph = self.alph * (1 - opt.mu * self.alph)
        if pr.comm.world > 1:
            pr.comm.exchange_rows(
                pr.layout, list(pr.graph.edges()), [eng.theta], [dests]
            )
        ext.mix_rows(eng.theta, rbuf, offs, idx, w, self.theta_next)
        eng.theta, self.theta_next = self.theta_next, eng.theta

        want_tl = bool(getattr(pr, "track_tloss", False))
        if eng.fused_step_available():
            off = eng.fused_advance()
            lb = eng.run_fused_mnist(off=off, want_loss=want_tl)
            eng.reduce_fused_grad()
        else:
            xb, yb = eng.next_batch()
            lb = eng.fwd_bwd(xb, yb, want_loss=want_tl)
        if want_tl and lb is not None:
            eng.update_tloss(lb)
        ext.axpy(eng.theta, eng.grad, -self.alph, True)
        eng.grad_is_zero = True

    def _opt_state(self):
        return {"alph": self.alph}

    def _load_opt_state(self, st):
        self.alph = st["alph"]

    def run(self, profiler=None):
        _run_rounds(self, profiler)


class DSGTStackedDriver:
    """DSGT round with fused (p, y) mixing kernels.

    Params and tracker ride one batched exchange (two rows per edge =
    the algorithm's 2x comm volume)."""

    def __init__(self, dsgt, pr):
        self.opt = dsgt
        self.pr = pr
        self.eng: StackedEngine = pr.stacked

    def prepare(self):
        eng = self.eng
        self.alpha = self.opt.alpha
        self.y = torch.zeros_like(eng.theta)
        self.g = torch.zeros_like(eng.theta)
        self.y_mix = torch.zeros_like(eng.theta)
        self.theta_next = torch.empty_like(eng.theta)
        self._plans = {}
        if self.opt.conf["init_grads"]:
            if eng.fused_step_available():
                eng.run_fused_mnist(off=eng.fused_advance())
                eng.reduce_fused_grad()
            else:
                xb, yb = eng.next_batch()
                eng.fwd_bwd(xb, yb)
            self.y.copy_(eng.grad)
            self.g.copy_(eng.grad)

    def _round_plan(self):
        from ..utils import graph_generation

        key = _edge_key(self.pr)
        plan = self._plans.get(key)
        if plan is not None:
            return plan
        eng = self.eng
        W = graph_generation.get_metropolis(self.pr.graph)
        remote_nodes, rbuf, _ = eng.remote_plan(width_factor=2)
        # destination views: [p | y] halves of each remote bundle row
        n = eng.n
        dests_p = {j: rbuf[r, :n] for r, j in enumerate(remote_nodes)} \
            if rbuf is not None else {}
        dests_y = {j: rbuf[r, n:] for r, j in enumerate(remote_nodes)} \
            if rbuf is not None else {}
        row_of = eng.row_map(remote_nodes)
        offs, idx, w = eng.build_csr(row_of, include_self=True, W=W)
        plan = (rbuf, dests_p, dests_y, offs, idx, w)
        if len(self._plans) < 256:
            self._plans[key] = plan
        return plan

    def step_round(self, k):
        opt, pr, eng = self.opt, self.pr, self.eng
        ext = eng.ext
        with _timer("update_graph"):
            pr.update_graph()
        with _timer("round_plan"):
            rbuf, dests_p, dests_y, offs, idx, w = self._round_plan()
        with _timer("exchange"):
            if pr.comm.world > 1:
                pr.comm.exchange_rows(
                    pr.layout, list(pr.graph.edges()),
                    [eng.theta, self.y], [dests_p, dests_y],
                )
        with _timer("mix"):
            ext.dsgt_mix(
                eng.theta, self.y, rbuf, offs, idx, w, self.theta_next,
                self.y_mix, self.alpha,
            )
            eng.theta, self.theta_next = self.theta_next, eng.theta

        want_tl = bool(getattr(pr, "track_tloss", False))
        if eng.fused_step_available():
            with _timer("train_step"):
                off = eng.fused_advance()
                lb = eng.run_fused_mnist(off=off, want_loss=want_tl)
                eng.reduce_fused_grad()
        else:
            with _timer("next_batch"):
                xb, yb = eng.next_batch()
            with _timer("fwd_bwd"):
                lb = eng.fwd_bwd(xb, yb, want_loss=want_tl)
        if want_tl and lb is not None:
            eng.update_tloss(lb)
        with _timer("y_update"):
            ext.dsgt_y_update(self.y_mix, eng.grad, self.g, self.y)

    def _opt_state(self):
        # y_mix is per-round scratch (overwritten by dsgt_mix before any
        # read) — excluded from checkpoints
        return {"y": self.y, "g": self.g}

    def _load_opt_state(self, st):
        self.y.copy_(st["y"])
        self.g.copy_(st["g"])

    def run(self, profiler=None):
        _run_rounds(self, profiler, skip_bootstrap_on_resume=True)


def _run_rounds(driver, profiler=None, skip_bootstrap_on_resume=False):
    """Shared outer loop for the stacked drivers: eval cadence,
    checkpoint/resume (optimizers/checkpointing.py), profiler hook."""
    from ..optimizers.checkpointing import load_checkpoint, save_checkpoint

    pr = driver.pr
    conf = driver.opt.conf
    eval_every = pr.conf["metrics_config"]["evaluate_frequency"]
    oits = conf["outer_iterations"]
    ck_every = conf.get("checkpoint_every", 0)
    ck_dir = conf.get("checkpoint_dir") or getattr(
        driver.opt, "checkpoint_dir", None
    )

    resume = conf.get("resume_from")
    if resume and skip_bootstrap_on_resume:
        # prepare() runs the DSGT init_grads bootstrap; suppress it,
        # the tracker state comes from the checkpoint
        saved = conf.get("init_grads")
        conf["init_grads"] = False
        driver.prepare()
        conf["init_grads"] = saved
    else:
        driver.prepare()
    k0 = 0
    if resume:
        k0, st = load_checkpoint(resume, pr)
        driver._load_opt_state(st)
    for k in range(k0, oits):
        if ck_every and k > k0 and k % ck_every == 0:
            save_checkpoint(ck_dir, pr, k - 1, driver._opt_state())
        if k % eval_every == 0 or k == oits - 1:
            pr.evaluate_metrics(at_end=(k == oits - 1))
        driver.step_round(k)
        if profiler is not None:
            profiler.step()
