"""CycleGAN trainer.

Replicates the reference trainer semantics (/root/reference/main.py:106-329)
on the MI355X execution model:

- 4 models (G: X->Y, F: Y->X, discriminators X, Y), 4 TF-style Adam
  optimizers (2e-4, beta1=0.5, beta2=0.9, eps=1e-7) over flat fp32 buffers;
- losses pre-scaled by 1/global_batch_size so SUM all-reduce over DP
  replicas is the exact global mean (main.py:172-174);
- per-step: one shared forward mega-graph; 4 gradient passes restricted to
  each model's variables (``torch.autograd.grad(loss, group.params)`` ==
  TF ``optimizer.minimize(var_list=...)``; grads land in the flat buffer
  via one batched multi-tensor copy), each followed immediately by an
  async RCCL all-reduce of that group's flat gradient, overlapping the next
  backward (reference runs the 4 groups serially, main.py:249-260);
- G's adversarial gradient flows through the frozen discriminator (the
  ``inputs=`` restriction limits accumulation, not the gradient path);
  each fake is discriminated ONCE and the node is shared by the
  adversarial and discriminator losses — gradient-identical to the
  reference's recomputed X(fake_x)/Y(fake_y) calls (main.py:239-245)
  under its var_list restriction, asserted by
  tests/test_trainer.py::test_train_step_grads_match_reference_structure;
- compute dtype bf16 on GPU (fp32 masters), fp32 on CPU.
"""

from __future__ import annotations

import os
from typing import Dict, Optional

import torch

from .models import Generator, Discriminator
from .ops import MAE, MSE, MSE_const, backend
from .ops.adam import FusedAdam
from .parallel import DistContext, GradSync, FlatParamGroup


class CycleGAN:
    LAMBDA_CYCLE = 10.0
    LAMBDA_IDENTITY = 0.5 * 10.0

    def __init__(self, args, ctx: DistContext):
        self.ctx = ctx
        self.device = ctx.device
        self.global_batch_size = args.global_batch_size
        self.compute_dtype = getattr(args, "compute_dtype", None) or (
            torch.bfloat16 if self.device.type == "cuda" else torch.float32)
        self._fp8 = bool(getattr(args, "fp8", False))
        if self._fp8:
            from .ops import set_fp8_mode
            set_fp8_mode(True)
            self.compute_dtype = torch.bfloat16

        self.checkpoint_dir = os.path.join(args.output_dir, "checkpoints")
        if ctx.is_main:
            os.makedirs(self.checkpoint_dir, exist_ok=True)
        self.checkpoint_path = os.path.join(self.checkpoint_dir, "checkpoint.pt")

        nrb = getattr(args, "num_residual_blocks", 9)
        self.G = Generator(num_residual_blocks=nrb).to(self.device)
        self.F = Generator(num_residual_blocks=nrb).to(self.device)
        self.X = Discriminator().to(self.device)
        self.Y = Discriminator().to(self.device)

        # mirror rank-0 init to every replica (reference S7 broadcast)
        for m in (self.G, self.F, self.X, self.Y):
            ctx.broadcast_module(m)

        self.groups = {name: FlatParamGroup(m) for name, m in
                       (("G", self.G), ("F", self.F), ("X", self.X), ("Y", self.Y))}
        self.optimizers = {name: FusedAdam(g.flat_param, g.flat_grad)
                           for name, g in self.groups.items()}
        self.sync = GradSync(ctx, timing=getattr(args, "verbose", 1) == 2)

        # shadow arenas: every bf16 compute form of a group refreshed by
        # ONE gather kernel after each optimizer step (ops/arena.py)
        self.arenas = {}
        if (self.device.type == "cuda"
                and self.compute_dtype == torch.bfloat16
                and backend.load_ext() is not None):
            from .ops.arena import ShadowArena
            from .ops import shadow as _shadow
            for name, m in (("G", self.G), ("F", self.F),
                            ("X", self.X), ("Y", self.Y)):
                a = ShadowArena(self.groups[name], m)
                _shadow.register_arena(a)
                self.arenas[name] = a

        # lazily-captured hip-graph step (main.py at N=1)
        self.graphed = None

    def _refresh_shadows(self):
        for a in self.arenas.values():
            a.refresh()

    # ---- loss functions (reference main.py:172-195) ----

    def reduce_mean(self, per_sample: torch.Tensor) -> torch.Tensor:
        return per_sample.sum() / self.global_batch_size

    def generator_loss(self, discriminate_fake):
        return self.reduce_mean(MSE_const(discriminate_fake, 1.0))

    def cycle_loss(self, real, cycled):
        return self.LAMBDA_CYCLE * self.reduce_mean(MAE(real, cycled))

    def identity_loss(self, real, same):
        return self.LAMBDA_IDENTITY * self.reduce_mean(MAE(real, same))

    def discriminator_loss(self, discriminate_real, discriminate_fake):
        real = MSE_const(discriminate_real, 1.0)
        fake = MSE_const(discriminate_fake, 0.0)
        return self.reduce_mean(0.5 * (real + fake))

    # ---- steps ----

    def _cast(self, t: torch.Tensor) -> torch.Tensor:
        return t.to(self.device, self.compute_dtype, non_blocking=True)

    _TRAIN_KEYS = ("loss_G/loss", "loss_G/cycle", "loss_G/identity",
                   "loss_G/total", "loss_F/loss", "loss_F/cycle",
                   "loss_F/identity", "loss_F/total", "loss_X/loss",
                   "loss_Y/loss")
    _TEST_KEYS = _TRAIN_KEYS + ("error/MAE(X, F(G(X)))",
                                "error/MAE(Y, G(F(Y)))",
                                "error/MAE(X, F(X))", "error/MAE(Y, G(Y))")

    def _zero_losses(self, keys) -> Dict[str, torch.Tensor]:
        z = torch.zeros((), device=self.device)
        return {k: z for k in keys}

    def _empty_train_step(self) -> Dict[str, torch.Tensor]:
        """Short final global batch under DP: this rank's slice is empty
        (the reference's MirroredStrategy replicas idle the same way,
        main.py:80-81). Contribute zero gradients but still join the
        all-reduces and take the identical optimizer step — after the
        SUM all-reduce every rank applies the same update, so replicas
        stay in sync."""
        for g in self.groups.values():
            g.flat_grad.zero_()
            self.sync.launch(g.flat_grad)
        self.sync.wait_all()
        for opt in self.optimizers.values():
            opt.step()
        for g in self.groups.values():
            g.bump_versions()
        self._refresh_shadows()
        return self._zero_losses(self._TRAIN_KEYS)

    def _forward_losses(self, x, y) -> Dict[str, torch.Tensor]:
        """The step's forward mega-graph: all 10 loss scalars, grads not
        yet taken (shared by the eager, fully-graphed, and segmented-graph
        step drivers)."""
        b = x.shape[0]
        # batched generator calls: every op is per-sample (convs, per-sample
        # InstanceNorm stats, per-sample losses), so G(cat(x,y)) is
        # numerically identical to G(x), G(y) — fewer, larger kernels.
        #
        # The cycle inputs are detached: under the reference's var_list
        # restriction (main.py:249-260) no optimizer ever follows the
        # fake_y -> G path out of F's cycle branch (F_cycle is only in
        # F_total, whose grads stop at F's own weights), so cutting it is
        # gradient-identical and lets F run as ONE 3b-batched call without
        # autograd descending a numerically-zero path into G.
        # torch.split, not slicing: split's backward is ONE cat of the
        # branch grads; slice backward is a zero-fill + copy + add per
        # branch (and leaves non-contiguous dy for the conv backwards)
        fake_y, same_y = self.G(torch.cat([x, y])).split(b)
        f_out = self.F(torch.cat([y, x, fake_y.detach()]))
        fake_x, same_x, cycle_x = f_out.split(b)
        cycle_y = self.G(fake_x.detach())

        # ONE discriminator pass per fake, shared by the adversarial and
        # discriminator losses (the reference recomputes X(fake_x) at
        # main.py:239-245 with identical values; TF graph-mode CSE merges
        # them the same way). X's own update never reaches fake_x: the
        # path stops at X's first conv weights.
        discriminate_fake_x = self.X(fake_x)
        discriminate_fake_y = self.Y(fake_y)

        G_loss = self.generator_loss(discriminate_fake_y)
        F_loss = self.generator_loss(discriminate_fake_x)
        G_cycle_loss = self.cycle_loss(y, cycle_y)
        F_cycle_loss = self.cycle_loss(x, cycle_x)
        G_identity_loss = self.identity_loss(y, same_y)
        F_identity_loss = self.identity_loss(x, same_x)
        return {
            "loss_G/loss": G_loss, "loss_G/cycle": G_cycle_loss,
            "loss_G/identity": G_identity_loss,
            "loss_G/total": G_loss + G_cycle_loss + G_identity_loss,
            "loss_F/loss": F_loss, "loss_F/cycle": F_cycle_loss,
            "loss_F/identity": F_identity_loss,
            "loss_F/total": F_loss + F_cycle_loss + F_identity_loss,
            "loss_X/loss": self.discriminator_loss(self.X(x),
                                                   discriminate_fake_x),
            "loss_Y/loss": self.discriminator_loss(self.Y(y),
                                                   discriminate_fake_y),
        }

    _GROUP_LOSS = (("G", "loss_G/total"), ("F", "loss_F/total"),
                   ("X", "loss_X/loss"), ("Y", "loss_Y/loss"))

    def _grads_for(self, losses: Dict[str, torch.Tensor], name: str,
                   retain: bool):
        """One group's backward pass into its flat grad buffer.

        torch.autograd.grad (not .backward): grads come back as fresh
        tensors and land in the flat buffer via ONE batched multi-tensor
        copy per group, skipping AccumulateGrad's per-param add into the
        .grad views (~300 tiny launches/step) and the flat zero-fill.
        retain_graph through X's pass: the shared discriminate_fake_*
        subgraphs are traversed by both the generator and discriminator
        passes."""
        key = dict(self._GROUP_LOSS)[name]
        self.groups[name].set_grads(
            torch.autograd.grad(losses[key], self.groups[name].params,
                                retain_graph=retain))

    def _optimize(self):
        for opt in self.optimizers.values():
            opt.step()
        for g in self.groups.values():
            g.bump_versions()  # invalidate bf16 shadow caches (see flat.py)
        self._refresh_shadows()
        if getattr(self, "_fp8", False):
            from .ops import fp8_state
            fp8_state.roll()  # delayed-scaling amax: prev <- cur

    def train_step(self, x, y) -> Dict[str, torch.Tensor]:
        x, y = self._cast(x), self._cast(y)
        if x.shape[0] == 0:
            return self._empty_train_step()
        losses = self._forward_losses(x, y)
        for i, (name, _) in enumerate(self._GROUP_LOSS):
            self._grads_for(losses, name, retain=i < 3)
            self.sync.launch(self.groups[name].flat_grad)
        self.sync.wait_all()
        self._optimize()
        G_loss = losses["loss_G/loss"]; G_cycle_loss = losses["loss_G/cycle"]
        G_identity_loss = losses["loss_G/identity"]; G_total = losses["loss_G/total"]
        F_loss = losses["loss_F/loss"]; F_cycle_loss = losses["loss_F/cycle"]
        F_identity_loss = losses["loss_F/identity"]; F_total = losses["loss_F/total"]
        X_loss = losses["loss_X/loss"]; Y_loss = losses["loss_Y/loss"]

        return {
            "loss_G/loss": G_loss.detach(), "loss_G/cycle": G_cycle_loss.detach(),
            "loss_G/identity": G_identity_loss.detach(), "loss_G/total": G_total.detach(),
            "loss_F/loss": F_loss.detach(), "loss_F/cycle": F_cycle_loss.detach(),
            "loss_F/identity": F_identity_loss.detach(), "loss_F/total": F_total.detach(),
            "loss_X/loss": X_loss.detach(), "loss_Y/loss": Y_loss.detach(),
        }

    @torch.no_grad()
    def cycle_step(self, x, y, training: bool = False):
        x, y = self._cast(x), self._cast(y)
        fake_y = self.G(x)
        cycle_x = self.F(fake_y)
        fake_x = self.F(y)
        cycle_y = self.G(fake_x)
        return fake_x, fake_y, cycle_x, cycle_y

    @torch.no_grad()
    def test_step(self, x, y) -> Dict[str, torch.Tensor]:
        x, y = self._cast(x), self._cast(y)
        if x.shape[0] == 0:  # empty DP slice of a short final batch
            return self._zero_losses(self._TEST_KEYS)
        fake_x, fake_y, cycle_x, cycle_y = self.cycle_step(x, y)

        discriminate_fake_x = self.X(fake_x)
        discriminate_fake_y = self.Y(fake_y)
        G_loss = self.generator_loss(discriminate_fake_y)
        F_loss = self.generator_loss(discriminate_fake_x)
        F_cycle_loss = self.cycle_loss(x, cycle_x)
        G_cycle_loss = self.cycle_loss(y, cycle_y)
        same_x = self.F(x)
        same_y = self.G(y)
        G_identity_loss = self.identity_loss(y, same_y)
        F_identity_loss = self.identity_loss(x, same_x)
        G_total = G_loss + G_cycle_loss + G_identity_loss
        F_total = F_loss + F_cycle_loss + F_identity_loss
        X_loss = self.discriminator_loss(self.X(x), discriminate_fake_x)
        Y_loss = self.discriminator_loss(self.Y(y), discriminate_fake_y)

        return {
            "loss_G/loss": G_loss, "loss_G/cycle": G_cycle_loss,
            "loss_G/identity": G_identity_loss, "loss_G/total": G_total,
            "loss_F/loss": F_loss, "loss_F/cycle": F_cycle_loss,
            "loss_F/identity": F_identity_loss, "loss_F/total": F_total,
            "loss_X/loss": X_loss, "loss_Y/loss": Y_loss,
            "error/MAE(X, F(G(X)))": self.reduce_mean(MAE(x, cycle_x)),
            "error/MAE(Y, G(F(Y)))": self.reduce_mean(MAE(y, cycle_y)),
            "error/MAE(X, F(X))": self.reduce_mean(MAE(x, same_x)),
            "error/MAE(Y, G(Y))": self.reduce_mean(MAE(y, same_y)),
        }

    # ---- checkpoint (reference main.py:148-170: one overwriting prefix,
    # all 4 models + 4 optimizer states, auto-resume if present) ----

    def save_checkpoint(self):
        if self.ctx.is_main:
            state = {
                "G": self.G.state_dict(), "F": self.F.state_dict(),
                "X": self.X.state_dict(), "Y": self.Y.state_dict(),
                "G_optimizer": self.optimizers["G"].state_dict(),
                "F_optimizer": self.optimizers["F"].state_dict(),
                "X_optimizer": self.optimizers["X"].state_dict(),
                "Y_optimizer": self.optimizers["Y"].state_dict(),
            }
            tmp = self.checkpoint_path + ".tmp"
            torch.save(state, tmp)
            os.replace(tmp, self.checkpoint_path)
            print(f"\nsaved checkpoint to {self.checkpoint_path}\n")
        self.ctx.barrier()

    def load_checkpoint(self) -> bool:
        if not os.path.exists(self.checkpoint_path):
            return False
        state = torch.load(self.checkpoint_path, map_location=self.device,
                           weights_only=True)
        self.G.load_state_dict(state["G"])
        self.F.load_state_dict(state["F"])
        self.X.load_state_dict(state["X"])
        self.Y.load_state_dict(state["Y"])
        for name in ("G", "F", "X", "Y"):
            self.optimizers[name].load_state_dict(state[f"{name}_optimizer"])
        self._refresh_shadows()
        print(f"\nloaded checkpoint from {self.checkpoint_path}\n")
        return True

    # ---- epoch-level metric reduction (S5, deferred to epoch end) ----

    def reduce_results(self, accumulated: Dict[str, list]) -> Dict[str, float]:
        """Mean over steps, SUM all-reduce over replicas, to host floats."""
        if not accumulated:
            return {}
        keys = list(accumulated.keys())
        means = torch.stack([torch.stack([v.float() for v in accumulated[k]]).mean()
                             for k in keys])
        self.ctx.all_reduce_(means)
        vals = means.tolist()
        return dict(zip(keys, vals))


class GraphedStep:
    """The steady-state train step captured as ONE hip graph.

    The whole step — bf16 shadow recasts, batched G/F/D forwards, the four
    backward passes, flat-grad copies and the four fused Adam updates —
    replays as a single graph launch, eliminating the per-kernel host
    launch gaps of ~300 small launches (the reference leans on TF's
    tracing compiler for the same effect, /root/reference/main.py:198-205;
    the MI355X-native equivalent is hipGraph capture).

    Requirements: static shapes, CUDA device, world_size == 1 (RCCL graph
    capture is intentionally not enabled yet). The optimizer's bias
    correction is read from a device scalar refreshed before every replay,
    so the captured step tracks t exactly like the eager path.
    """

    def __init__(self, gan: CycleGAN, x: torch.Tensor, y: torch.Tensor,
                 warmup: int = 2, preserve_state: bool = False):
        # world_size > 1 capture (RCCL collectives inside the graph) is
        # proven at 1 rank (profiles/rccl_shakeout.md) but intentionally
        # opt-in for multi-rank until measured there: CYG_GRAPH_DIST=1.
        assert gan.ctx.device.type == "cuda" and (
            gan.ctx.world_size == 1
            or os.environ.get("CYG_GRAPH_DIST") == "1")
        self.gan = gan
        self.sx = gan._cast(x).clone()
        self.sy = gan._cast(y).clone()
        # preserve_state: roll model/optimizer state back after the warmup
        # steps so capturing mid-training perturbs nothing (main.py uses
        # this to capture on the first batch of an epoch)
        saved = None
        if preserve_state:
            saved = {n: (g.flat_param.clone(), gan.optimizers[n].m.clone(),
                         gan.optimizers[n].v.clone(), gan.optimizers[n].t)
                     for n, g in gan.groups.items()}
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup):
                gan.train_step(self.sx, self.sy)
        torch.cuda.current_stream().wait_stream(s)
        for opt in gan.optimizers.values():
            opt.prepare_graph()
        # capture records the kernel sequence without executing it: the
        # optimizer step count is NOT advanced here, so replay #1 runs the
        # exact t the eager path would
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out = gan.train_step(self.sx, self.sy)
            self._packed = torch.stack([self.out[k] for k in self.gan._TRAIN_KEYS])
        if saved is not None:
            with torch.no_grad():
                for n, (p, m, v, t) in saved.items():
                    gan.groups[n].flat_param.copy_(p)
                    gan.optimizers[n].m.copy_(m)
                    gan.optimizers[n].v.copy_(v)
                    gan.optimizers[n].t = t
                gan._refresh_shadows()

    def __call__(self, x=None, y=None) -> Dict[str, torch.Tensor]:
        if x is not None:
            self.sx.copy_(self.gan._cast(x))
            self.sy.copy_(self.gan._cast(y))
        for opt in self.gan.optimizers.values():
            opt.advance_lr()
        self.graph.replay()
        for g in self.gan.groups.values():
            g.bump_versions()
        return self.out

    def call_cloned(self, x, y) -> Dict[str, torch.Tensor]:
        """Replay and return loss scalars detached from the static graph
        outputs (ONE clone kernel), safe to accumulate across steps."""
        self(x, y)
        p = self._packed.clone()
        return {k: p[i] for i, k in enumerate(self.gan._TRAIN_KEYS)}


class SegmentedGraphedStep:
    """Multi-rank graph step: the step is captured as FIVE RCCL-free hip
    graphs (forward + G-grads, F-grads, X-grads, Y-grads, optimizers)
    sharing one capture pool, and the four flat-gradient all-reduces are
    issued EAGERLY between the replays. Multi-rank runs get the captured
    kernel-launch profile (eager measured ~3.5% slower at N=1) without
    capturing any collective — RCCL-in-graph capture stays opt-in
    (CYG_GRAPH_DIST=1, GraphedStep). Collective order matches the eager
    path exactly (G, F, X, Y), so graphed and eager/empty-slice ranks
    interoperate."""

    def __init__(self, gan: CycleGAN, x: torch.Tensor, y: torch.Tensor,
                 warmup: int = 2, preserve_state: bool = False):
        assert gan.ctx.device.type == "cuda"
        self.gan = gan
        self.sx = gan._cast(x).clone()
        self.sy = gan._cast(y).clone()
        saved = None
        if preserve_state:
            saved = {n: (g.flat_param.clone(), gan.optimizers[n].m.clone(),
                         gan.optimizers[n].v.clone(), gan.optimizers[n].t)
                     for n, g in gan.groups.items()}
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup):
                gan.train_step(self.sx, self.sy)
        torch.cuda.current_stream().wait_stream(s)
        for opt in gan.optimizers.values():
            opt.prepare_graph()

        self.names = [n for n, _ in gan._GROUP_LOSS]
        self.graphs = []
        g1 = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g1):
            losses = gan._forward_losses(self.sx, self.sy)
            gan._grads_for(losses, "G", retain=True)
        self.graphs.append(g1)
        pool = g1.pool()
        for i, name in enumerate(self.names[1:], start=1):
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=pool):
                gan._grads_for(losses, name, retain=i < 3)
            self.graphs.append(g)
        gopt = torch.cuda.CUDAGraph()
        with torch.cuda.graph(gopt, pool=pool):
            for opt in gan.optimizers.values():
                opt.step()
            gan._refresh_shadows()
            if getattr(gan, "_fp8", False):
                from .ops import fp8_state
                fp8_state.roll()
            self._packed = torch.stack([losses[k] for k in gan._TRAIN_KEYS])
        self.graphs.append(gopt)
        self.out = {k: losses[k].detach() for k in gan._TRAIN_KEYS}

        if saved is not None:
            with torch.no_grad():
                for n, (p, m, v, t) in saved.items():
                    gan.groups[n].flat_param.copy_(p)
                    gan.optimizers[n].m.copy_(m)
                    gan.optimizers[n].v.copy_(v)
                    gan.optimizers[n].t = t
                gan._refresh_shadows()

    def __call__(self, x=None, y=None) -> Dict[str, torch.Tensor]:
        gan = self.gan
        if x is not None:
            self.sx.copy_(gan._cast(x))
            self.sy.copy_(gan._cast(y))
        for opt in gan.optimizers.values():
            opt.advance_lr()
        for i, name in enumerate(self.names):
            self.graphs[i].replay()
            gan.sync.launch(gan.groups[name].flat_grad)
        gan.sync.wait_all()
        self.graphs[4].replay()
        for g in gan.groups.values():
            g.bump_versions()
        return self.out

    def call_cloned(self, x, y) -> Dict[str, torch.Tensor]:
        self(x, y)
        p = self._packed.clone()
        return {k: p[i] for i, k in enumerate(self.gan._TRAIN_KEYS)}
