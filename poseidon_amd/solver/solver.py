"""Solvers: SGD / Nesterov / AdaGrad with the distributed training loop.

Parity: /root/reference/src/caffe/solver.cpp --
  Solve loop (:246-402), lr policies (:767-790), momentum/decay update rules
  (:815-892 SGD, :1013-1120 Nesterov, :1240-1364 AdaGrad -- fused into one
  HIP kernel each, ops.functional.sgd_update etc.), Snapshot (:632-667),
  Restore (:670-696), Test (:552-628), PrintNetOutputs (:699-756).

DWBP (:405-451) is realized as stream/event scheduling: Net.backward fires a
callback per learnable layer; GradReducer all-reduces that layer's grads on
a side HIP stream while backprop continues. SFB (:477-531) goes through
solver/sfb.py. The PS Clock/GlobalBarrier protocol collapses to BSP
(staleness=0, the reference's own recommended setting, docs/performance.md).
"""

from __future__ import annotations

import os
import time
from typing import Dict, List, Optional

import torch

from ..core.context import ctx
from ..core.net import Net, TRAIN, TEST
from ..core.blob import Blob
from ..ops import functional as ops
from ..parallel import comm
from ..proto import (Message, read_proto_text, read_proto_binary,
                     write_proto_binary)
from .sfb import SFBReducer, enable_sfb


def get_solver(param: Message, **kw) -> "SGDSolver":
    t = param.enum_name("solver_type")
    return {"SGD": SGDSolver, "NESTEROV": NesterovSolver,
            "ADAGRAD": AdaGradSolver}[t](param, **kw)


class SGDSolver:
    def __init__(self, param: Message, use_sfb: Optional[bool] = None,
                 verbose: Optional[bool] = None):
        self.param = param
        c = ctx()
        self.verbose = c.is_root() if verbose is None else verbose
        self.iter = 0

        net_param = self._load_net_param(param)
        train_state = Message("NetState", phase=TRAIN)
        if param.has("train_state"):
            train_state.merge_from(param.train_state)
            train_state.phase = TRAIN
        self.net = Net(net_param, phase=TRAIN, state=train_state)

        # test nets (solver.cpp InitTestNets, simplified: net + test_state's)
        self.test_nets: List[Net] = []
        test_states = list(param.test_state)
        n_tests = len(list(param.test_iter))
        for ti in range(n_tests):
            st = Message("NetState", phase=TEST)
            if ti < len(test_states):
                st.merge_from(test_states[ti])
                st.phase = TEST
            tn = Net(net_param, phase=TEST, state=st)
            self._share_params(tn)
            self.test_nets.append(tn)

        # optimizer state
        self.history: Dict[int, torch.Tensor] = {}
        for i, ps in enumerate(self.net.params):
            if ps.owner == i and ps.lr_mult != 0.0:
                self.history[i] = torch.zeros_like(ps.blob.diff)

        # distributed
        self.distributed = comm.init_distributed()
        if self.distributed:
            comm.broadcast_params([ps for i, ps in enumerate(self.net.params)
                                   if ps.owner == i])
        self.reducer = comm.GradReducer(self.net) if self.distributed else None
        if not self.distributed and c.device == "cuda":
            # single-GPU: conv wgrads stay in their khwc scratch through
            # backward and ONE unpack_mt kernel folds them into the NCHW
            # diffs at the end (net._flush_deferred_unpacks). Multi-rank
            # keeps per-layer unpack: DWBP all-reduces each layer's final
            # grad as soon as its backward completes.
            for l in self.net.layers:
                if l.type_name == "CONVOLUTION":
                    l.defer_unpack = True
        want_sfb = c.use_sfb if use_sfb is None else use_sfb
        self.sfb = None
        if self.distributed and want_sfb:
            marked = enable_sfb(self.net, c.world_size)
            if marked:
                self.sfb = SFBReducer(
                    marked,
                    self.reducer.comm_stream if self.reducer.use_stream else None)
                if self.verbose:
                    print(f"[poseidon] SFB active on: "
                          f"{[l.name for l in marked]}", flush=True)

        self._net_outputs_rows: List[List[float]] = []
        self._net_outputs_cols: List[str] = []
        self._t0 = time.time()

        # hipGraph capture of the steady-state iteration (HIP streams and
        # graphs instead of a tracing compiler): opt-in via enable_graph().
        self._use_graph = False
        self._graph = None
        self._graph_loss = None
        self._lr_dev: Optional[torch.Tensor] = None
        # multi-tensor update table (one kernel for ALL params); keyed on
        # tensor identities so restore()/load_weights() invalidate it
        self._mt = None

    # ------------------------------------------------------------------
    # hipGraph iteration capture
    # ------------------------------------------------------------------
    def enable_graph(self) -> bool:
        """Capture fwd+bwd+update as one hipGraph and replay per iteration
        (collapses per-kernel launch gaps -- GoogLeNet's hundreds of small
        kernels). Requires: GPU, SGD solver, static data layers (constant
        DummyData), L2 regularization. Works under multi-rank DP too: RCCL
        collectives are hipGraph-capturable, so the DWBP all-reduces and SFB
        all-gathers replay inside the graph; capture success is agreed
        across ranks (all-reduce MIN) so no rank replays while another runs
        eager. Returns True if enabled."""
        c = ctx()
        if c.device != "cuda" or type(self) is not SGDSolver:
            return False
        if self.param.regularization_type == "L1":
            return False
        if self.distributed:
            import torch.distributed as dist
            if dist.get_backend() != "nccl":
                return False  # gloo collectives are host-side: not capturable
        for layer in self.net.layers:
            t = layer.type_name
            if t in ("DATA", "IMAGE_DATA", "MEMORY_DATA", "WINDOW_DATA"):
                return False  # per-iter host work cannot replay
            if t == "DUMMY_DATA" and any(layer.refill):
                layer.forward(self.net.bottoms[self.net.layers.index(layer)],
                              self.net.tops[self.net.layers.index(layer)])
                layer.refill = [False] * len(layer.refill)  # freeze fills
        self._use_graph = True
        return True

    def _graph_body(self) -> torch.Tensor:
        loss = self.forward_backward()
        if not self._mt_update(0.0, lr_dev=self._lr_dev):
            for i, ps in enumerate(self.net.params):
                if ps.owner == i and ps.lr_mult != 0.0:
                    wd = float(self.param.weight_decay or 0.0) * ps.decay_mult
                    if self.distributed:
                        wd *= ctx().world_size
                    ops.sgd_update(ps.blob.data, ps.blob.diff,
                                   self.history[i], ps.lr_mult,
                                   float(self.param.momentum or 0.0),
                                   wd, lr_dev=self._lr_dev)
        return loss

    # -- multi-tensor update: one sgd_mt kernel covers every param ------
    def _mt_update(self, rate: float, lr_dev=None) -> bool:
        """Returns True if the whole update ran as one multi-tensor kernel
        (SGD + L2 + GPU only; Nesterov/AdaGrad subclasses and the L1 path
        keep per-param kernels). The descriptor table is rebuilt whenever
        any tensor identity changes (restore, first backward, ...)."""
        if type(self) is not SGDSolver or ctx().device != "cuda":
            return False
        if self.param.regularization_type == "L1":
            return False
        items = [(i, ps) for i, ps in enumerate(self.net.params)
                 if ps.owner == i and ps.lr_mult != 0.0]
        if not items or not all(ps.blob.has_diff() for _, ps in items):
            return False
        key = [(id(ps.blob.data), id(ps.blob.diff), id(self.history[i]))
               for i, ps in items]
        if self._mt is None or self._mt[1] != key:
            wd0 = float(self.param.weight_decay or 0.0)
            if self.distributed:
                wd0 *= ctx().world_size
            mt = ops.sgd_mt_prepare(
                [ps.blob.data for _, ps in items],
                [ps.blob.diff for _, ps in items],
                [self.history[i] for i, _ in items],
                [ps.lr_mult for _, ps in items],
                [wd0 * ps.decay_mult for _, ps in items])
            self._mt = (mt, key)
        ops.sgd_mt_run(self._mt[0], rate,
                       float(self.param.momentum or 0.0), lr_dev)
        return True

    def _capture_graph(self) -> None:
        dev = ctx().torch_device
        self._lr_dev = torch.zeros(1, dtype=torch.float32, device=dev)
        self._lr_dev.fill_(self.get_learning_rate())
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            # Allocation + identity warmup (inits RCCL comms too). THREE
            # iterations before capture: the multi-tensor tables settle at
            # iter 2 (zero table: built at 0 without the conv dwk scratch,
            # invalidated at 1 when backward allocates it, rebuilt at 2) and
            # a table rebuild is a pageable H2D copy -- forbidden inside
            # capture.
            for _ in range(3):
                self._graph_body()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        ok = True
        err: Optional[Exception] = None
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._graph_loss = self._graph_body()
        except Exception as e:  # noqa: BLE001
            ok, err = False, e
            # capture never executes collectives, only records them, so a
            # mismatched record count across ranks cannot deadlock here
            torch.cuda.synchronize()
        if self.distributed:
            # all ranks must agree before anyone replays: a rank replaying a
            # graphed all-reduce against a rank running eager would hang
            import torch.distributed as dist
            flag = torch.tensor([1.0 if ok else 0.0], device=dev)
            dist.all_reduce(flag, op=dist.ReduceOp.MIN)
            ok = ok and float(flag.item()) > 0.0
        if not ok:
            raise RuntimeError(f"graph capture failed on some rank: {err}")
        self._graph = g

    def _step_graphed(self, iters: int) -> float:
        p = self.param
        last_loss = 0.0
        if self._graph is None:
            self._capture_graph()
        while iters > 0:
            self._lr_dev.fill_(self.get_learning_rate())
            self._graph.replay()
            if p.display and self.iter % p.display == 0:
                last_loss = float(self._graph_loss.item())
                self._display(last_loss, self.get_learning_rate())
            self.iter += 1
            iters -= 1
        return last_loss

    # ------------------------------------------------------------------
    @staticmethod
    def _load_net_param(param: Message) -> Message:
        if param.has("net_param"):
            return param.net_param
        if param.has("train_net_param"):
            return param.train_net_param
        path = param.net if param.has("net") else param.train_net
        if not path:
            raise ValueError("solver specifies no train net")
        return read_proto_text(path, "NetParameter")

    def _share_params(self, other: Net) -> None:
        by_name = {l.name: l for l in self.net.layers}
        for layer in other.layers:
            src = by_name.get(layer.name)
            if src is None or not src.blobs:
                continue
            for pb, sb in zip(layer.blobs, src.blobs):
                pb.share_data(sb)

    # -- learning rate policies (solver.cpp:767-790) --------------------
    def get_learning_rate(self) -> float:
        p = self.param
        policy = p.lr_policy or "fixed"
        base = float(p.base_lr)
        if policy == "fixed":
            return base
        if policy == "step":
            return base * (p.gamma ** (self.iter // p.stepsize))
        if policy == "exp":
            return base * (p.gamma ** self.iter)
        if policy == "inv":
            return base * (1.0 + p.gamma * self.iter) ** (-p.power)
        if policy == "poly":
            return base * (1.0 - self.iter / float(p.max_iter)) ** p.power
        raise ValueError(f"unknown lr policy {policy!r}")

    # -- one fused update per param -------------------------------------
    def _apply_update(self, idx: int, ps, rate: float) -> None:
        local_rate = rate * ps.lr_mult
        wd = float(self.param.weight_decay or 0.0) * ps.decay_mult
        # PS semantics sum per-worker decay terms (each worker adds λW before
        # pushing) -- fold into the post-reduce update:
        if self.distributed:
            wd *= ctx().world_size
        mom = float(self.param.momentum or 0.0)
        reg = self.param.regularization_type
        grad = ps.blob.diff
        if reg == "L1" and wd != 0.0:
            grad = grad + wd * torch.sign(ps.blob.data)
            wd = 0.0
        ops.sgd_update(ps.blob.data, grad, self.history[idx],
                       local_rate, mom, wd)

    # ------------------------------------------------------------------
    def forward_backward(self) -> torch.Tensor:
        net = self.net
        net.zero_param_diffs()
        loss = net.forward_async()
        if self.reducer:
            self.reducer.begin_iter()

            def cb(layer_idx, layer):
                if self.sfb is not None:
                    self.sfb.on_layer_done(layer)
                self.reducer.on_layer_done(layer_idx, layer)
            net.backward(post_layer_cb=cb)
            self.reducer.flush()
            self.reducer.wait()
            if self.sfb is not None:
                self.sfb.finish()
        else:
            net.backward()
        return loss

    def step(self, iters: int) -> float:
        if self._use_graph:
            try:
                return self._step_graphed(iters)
            except Exception as e:  # noqa: BLE001 -- capture unsupported
                import sys
                print(f"[poseidon] graph capture failed ({e}); "
                      "falling back to eager", file=sys.stderr, flush=True)
                self._use_graph = False
                self._graph = None
        last_loss = 0.0
        p = self.param
        while iters > 0:
            if (p.snapshot and self.iter > 0
                    and self.iter % p.snapshot == 0):
                self.snapshot()
            if (p.test_interval and self.iter % p.test_interval == 0
                    and (self.iter > 0 or bool(p.test_initialization))):
                self.test_all()
            loss_t = self.forward_backward()
            rate = self.get_learning_rate()
            if not self._mt_update(rate):
                for i, ps in enumerate(self.net.params):
                    if ps.owner == i and ps.lr_mult != 0.0:
                        self._apply_update(i, ps, rate)
            if p.display and self.iter % p.display == 0:
                last_loss = float(loss_t.item())
                self._display(last_loss, rate)
            self.iter += 1
            iters -= 1
        return last_loss

    def solve(self, resume_file: Optional[str] = None) -> None:
        p = self.param
        if resume_file:
            self.restore(resume_file)
        comm.barrier()
        self.step(int(p.max_iter) - self.iter)
        if bool(p.snapshot_after_train):
            self.snapshot()
        if p.test_interval:
            self.test_all()
        comm.barrier()

    # ------------------------------------------------------------------
    def _display(self, loss: float, rate: float) -> None:
        c = ctx()
        # Flatten WHOLE output blobs into the row (reference writes every
        # element of every net output into the net-output PS table,
        # solver.cpp:336-366) -- vector outputs get name_<j> columns.
        outs = []
        for name in sorted(self.net.output_blob_names):
            b = self.net.blobs[name]
            flat = b.data.detach().reshape(-1).to(torch.float64).cpu()
            if flat.numel() == 1:
                outs.append((name, float(flat[0])))
            else:
                for j in range(flat.numel()):
                    outs.append((f"{name}_{j}", float(flat[j])))
        vals = torch.tensor([loss] + [v for _, v in outs], dtype=torch.float64)
        if self.distributed:
            comm.allreduce_metrics(vals)
            vals /= c.world_size
        if not self._net_outputs_cols:
            self._net_outputs_cols = ["iter", "time", "loss"] + [n for n, _ in outs]
        row = [float(self.iter), time.time() - self._t0] +             [float(v) for v in vals]
        self._net_outputs_rows.append(row)
        if self.verbose:
            extra = " ".join(f"{n}={float(v):.4f}"
                             for (n, _), v in zip(outs, vals[1:]))
            print(f"[poseidon] iter {self.iter} loss {vals[0]:.6f} "
                  f"lr {rate:.6g} {extra}", flush=True)

    def test_all(self) -> List[Dict[str, float]]:
        results = []
        for ti, tn in enumerate(self.test_nets):
            results.append(self.test(ti))
        return results

    def test(self, test_net_id: int = 0) -> Dict[str, float]:
        """Divide test_iter across ranks (solver.cpp:552-628 divides across
        clients x threads); aggregate outputs with an all-reduce."""
        tn = self.test_nets[test_net_id]
        total_iters = int(list(self.param.test_iter)[test_net_id])
        c = ctx()
        my_iters = total_iters // c.world_size
        if c.rank < total_iters % c.world_size:
            my_iters += 1
        # Metric names come from the net topology, NOT from which ranks ran
        # iterations: a rank with my_iters == 0 (world_size > test_iter)
        # must still contribute an identically-shaped zero tensor to the
        # all-reduce or the collective hangs/corrupts.
        names = sorted(tn.output_blob_names)
        sums: Dict[str, float] = {n: 0.0 for n in names}
        for _ in range(my_iters):
            tn.forward()
            for name in names:
                b = tn.blobs[name]
                sums[name] = sums.get(name, 0.0) + float(b.data.sum().item())
        vals = torch.tensor([sums[n] for n in names] or [0.0],
                            dtype=torch.float64)
        if self.distributed:
            comm.allreduce_metrics(vals)
        out: Dict[str, float] = {}
        for i, n in enumerate(names):
            avg = float(vals[i]) / total_iters
            if self.verbose:
                print(f"[poseidon] test net {test_net_id} {n} = {avg:.5f}",
                      flush=True)
            out[n] = avg
        return out

    # -- checkpoint / resume (solver.cpp:632-696) ------------------------
    def snapshot(self) -> str:
        prefix = self.param.snapshot_prefix or "poseidon"
        model_path = f"{prefix}_iter_{self.iter}.caffemodel"
        c = ctx()
        if c.is_root():
            net_proto = self.net.to_proto()
            gid = 0
            for lp in net_proto.layers:
                for b in lp.blobs:
                    b.blob_mode = "GLOBAL"
                    b.global_id = gid
                    gid += 1
            write_proto_binary(net_proto, model_path)
        state = Message("SolverState", iter=self.iter, learned_net=model_path)
        for i, ps in enumerate(self.net.params):
            if ps.owner == i and i in self.history:
                h = self.history[i]
                hb = Blob(tuple(h.shape), device=h.device)
                hb.data = h
                state.history.append(hb.to_proto())
        state_path = f"{prefix}_iter_{self.iter}.solverstate.{c.rank}.0"
        write_proto_binary(state, state_path)
        if self.verbose:
            print(f"[poseidon] snapshot -> {model_path}", flush=True)
        return model_path

    def restore(self, state_file: str) -> None:
        c = ctx()
        if not os.path.exists(state_file):
            # Resolve the conventional suffix-less path first: the snapshot
            # writer emits '<prefix>_iter_N.solverstate.<rank>.0', so try
            # appending our rank / rank 0 before assuming state_file already
            # carries a numeric suffix (reference falls back to thread 0,
            # solver.cpp:670-696).
            for cand in (f"{state_file}.{c.rank}.0", f"{state_file}.0.0",
                         f"{state_file.rsplit('.', 2)[0]}.0.0"):
                if os.path.exists(cand):
                    state_file = cand
                    break
        state = read_proto_binary(state_file, "SolverState")
        self.iter = int(state.iter)
        hist_protos = list(state.history)
        own = [i for i, ps in enumerate(self.net.params)
               if ps.owner == i and i in self.history]
        # History blobs pair with owned params positionally (the reference
        # keys by order too) -- but refuse a silent mispairing when the
        # prototxt changed between save and load (solver.cpp:1004-1008).
        if len(hist_protos) != len(own):
            raise ValueError(
                f"solverstate carries {len(hist_protos)} history blobs but "
                f"net has {len(own)} learnable params -- prototxt mismatch?")
        for i, hp in zip(own, hist_protos):
            want = self.history[i]
            if hp.num * hp.channels * hp.height * hp.width != want.numel():
                raise ValueError(
                    f"history blob for param {i} has "
                    f"{hp.num * hp.channels * hp.height * hp.width} elements, "
                    f"net param has {want.numel()}")
            hb = Blob((), device=want.device)
            hb.from_proto(hp)
            self.history[i] = hb.data.view(want.shape).to(want.dtype)
        if state.has("learned_net") and os.path.exists(state.learned_net):
            net_proto = read_proto_binary(state.learned_net, "NetParameter")
            self.net.copy_trained_layers_from(net_proto)
        if self.distributed:
            comm.broadcast_params([ps for i, ps in enumerate(self.net.params)
                                   if ps.owner == i])

    def load_weights(self, weights_file: str) -> None:
        """Finetuning entry (--weights, caffe_engine.cpp:277-282)."""
        net_proto = read_proto_binary(weights_file, "NetParameter")
        self.net.copy_trained_layers_from(net_proto)
        if self.distributed:
            comm.broadcast_params([ps for i, ps in enumerate(self.net.params)
                                   if ps.owner == i])

    def write_net_outputs(self, path: str) -> None:
        """CSV of display-time metrics (PrintNetOutputs, solver.cpp:699-756)."""
        if not ctx().is_root():
            return
        cols = self._net_outputs_cols or ["iter", "time", "loss"]
        with open(path + ".netoutputs", "w") as f:
            f.write(",".join(cols) + "\n")
            for row in self._net_outputs_rows:
                f.write(",".join(f"{v:.6f}" for v in row) + "\n")


class NesterovSolver(SGDSolver):
    def _apply_update(self, idx: int, ps, rate: float) -> None:
        local_rate = rate * ps.lr_mult
        wd = float(self.param.weight_decay or 0.0) * ps.decay_mult
        if self.distributed:
            wd *= ctx().world_size
        mom = float(self.param.momentum or 0.0)
        ops.nesterov_update(ps.blob.data, ps.blob.diff, self.history[idx],
                            local_rate, mom, wd)


class AdaGradSolver(SGDSolver):
    def _apply_update(self, idx: int, ps, rate: float) -> None:
        local_rate = rate * ps.lr_mult
        wd = float(self.param.weight_decay or 0.0) * ps.decay_mult
        if self.distributed:
            wd *= ctx().world_size
        ops.adagrad_update(ps.blob.data, ps.blob.diff, self.history[idx],
                           local_rate, float(self.param.delta), wd)
