"""Fused many-model client engine — the MI355X rounds/sec lever.

The reference simulates clients *sequentially* inside each Ray actor
(reference: actor.py:23-33).  Here the whole rank-local client population
trains as ONE batched computation:

* per-client parameters live in a single contiguous [C, d] fp32 HBM slab
  (``ParamSpec.batched_views`` exposes zero-copy [C, *shape] views);
* the per-client forward+backward is ``torch.func.vmap`` over
  ``torch.func.grad`` of a functional loss — convs become grouped convs over
  the client dim, so the GPU sees effective batch C×B instead of C separate
  batch-B launches;
* FedSGD (local_steps == 1, all clients start at θ) skips materializing the
  parameter slab: U = −lr·g directly, with params passed shared
  (``in_dims=None``) — the classic per-"sample" gradient trick with clients
  as the sample dim;
* built-in training-time attacks are fused: label-flip is an integer map on
  the stacked target tensor (K14), sign-flip is a per-client −1 gradient
  multiplier and tighter loss clamp (K13) — no per-client Python anywhere.

Loss clamp semantics match the reference: honest clients clamp loss to
[0, 1e6] (client.py:191), sign-flippers to [0, 1e5]
(signflippingclient.py:17).

BatchNorm: running-stats updates are per-client-meaningless in FL and
incompatible with vmap; ``make_vmap_safe`` switches BN modules to
batch-stats mode (track_running_stats=False).  Models built with
``norm="batch-local"`` or ``norm="group"`` pass through untouched.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn
from torch.func import functional_call, grad, vmap

from blades_amd.client import BladesClient
from blades_amd.engine.flat import ParamSpec

Tensor = torch.Tensor


def make_vmap_safe(model: nn.Module) -> nn.Module:
    """Switch every BatchNorm to batch-stats mode, in place; returns model."""
    for m in model.modules():
        if isinstance(m, nn.modules.batchnorm._BatchNorm):
            m.track_running_stats = False
            m.running_mean = None
            m.running_var = None
            m.num_batches_tracked = None
    return model


class FusedEngine:
    """Trains a set of fusable clients as one batched computation.

    Parameters
    ----------
    model : the global architecture (a template; weights come per round)
    spec : flat layout (must match the model)
    device : target device
    client_chunk : optional vmap chunk size over the client dim to bound
        activation memory (None = all clients at once)
    """

    def __init__(self, model: nn.Module, spec: ParamSpec, device,
                 client_chunk: Optional[int] = None):
        import copy

        self.device = torch.device(device)
        self.spec = spec
        self.base = make_vmap_safe(copy.deepcopy(model)).to(self.device)
        self.base.train()
        # buffers (none for vmap-safe BN; kept for generality) are shared
        self.buffers = {k: v.detach().clone() for k, v in self.base.named_buffers()}
        self.client_chunk = client_chunk
        # NOTE round 2: the direct-MFMA population-conv path (popconv) was
        # REMOVED after losing both of its target configs on hardware —
        # docs/popconv_postmortem.md has the measurements and reasoning.

    # ------------------------------------------------------------ internals
    def _loss(self, params_tuple: Tuple[Tensor, ...], x: Tensor, y: Tensor,
              clamp_hi: Tensor, grad_sign: Tensor) -> Tensor:
        params = dict(zip(self.spec.names, params_tuple))
        out = functional_call(self.base, (params, self.buffers), (x,))
        loss = self.base_loss(out, y)
        # reference loss clamp (client.py:191 / signflippingclient.py:17):
        # clamp to [0, hi]; min/minimum keep the same zero-gradient-outside
        # semantics torch.clamp has.  grad_sign folds sign-flip (K13) in:
        # grad(sign·loss) = sign·grad(loss).
        loss = torch.clamp(loss, min=0.0)
        loss = torch.minimum(loss, clamp_hi)
        return loss * grad_sign

    def _build_grad_fn(self, shared_params: bool):
        param_in = None if shared_params else 0
        g = grad(self._loss, argnums=0)
        return vmap(g, in_dims=((param_in,) * len(self.spec.names) if not shared_params else None,
                                0, 0, 0, 0),
                    chunk_size=self.client_chunk)

    # -------------------------------------------------------------- training
    base_loss = staticmethod(nn.functional.cross_entropy)

    @torch.no_grad()
    def _client_vectors(self, clients: Sequence[BladesClient]):
        key = tuple(id(c) for c in clients)
        cached = getattr(self, "_cv_cache", None)
        if cached is not None and cached[0] == key:
            return cached[1]
        result = self._client_vectors_uncached(clients)
        self._cv_cache = (key, result)
        return result

    @torch.no_grad()
    def _client_vectors_uncached(self, clients: Sequence[BladesClient]):
        C = len(clients)
        clamp_hi = torch.full((C,), 1e6, device=self.device)
        grad_sign = torch.ones((C,), device=self.device)
        tt = [None] * C
        for i, c in enumerate(clients):
            if hasattr(c, "fused_loss_clamp"):
                clamp_hi[i] = float(c.fused_loss_clamp)
            if hasattr(c, "fused_grad_sign"):
                grad_sign[i] = float(c.fused_grad_sign)
            if hasattr(c, "fused_target_transform"):
                tt[i] = c.fused_target_transform
        return clamp_hi, grad_sign, tt

    def _stack_step(self, dataset, clients, step_idx, batches) -> Tuple[Tensor, Tensor]:
        """Stack per-client batch ``step_idx`` into [C, B, ...] device tensors."""
        xs, ys = [], []
        for c in clients:
            x, y = batches[c.id()][step_idx]
            xs.append(x)
            ys.append(y)
        X = torch.stack(xs).to(self.device, non_blocking=True)
        Y = torch.stack(ys).to(self.device, non_blocking=True)
        return X, Y

    def run_round(self, theta: Tensor, clients: List[BladesClient], dataset,
                  local_steps: int, lr,
                  out: Optional[Tensor] = None, data=None) -> Tensor:
        """Returns the update slab U = θ_after − θ [C, d] on ``self.device``.

        ``theta`` is the flat global parameter vector (device-resident).
        ``out``: optional [C, d] destination (may be a padded-row view) the
        updates are written into directly — saves a slab copy per round.
        ``lr`` may be a float or a 0-dim device tensor (hipGraph capture
        keeps learning rates in device scalars so schedulers need no
        re-capture).  ``data``: pre-staged [(X [C,B,...], Y [C,B])] per
        local step — bypasses the dataset (the graph path owns static
        buffers).
        """
        C = len(clients)
        clamp_hi, grad_sign, target_tfms = self._client_vectors(clients)

        if data is not None:
            steps_data = data
        else:
            # fetch all batches up front (device datasets return views)
            stacked = getattr(dataset, "get_stacked_train_data", None)
            if stacked is not None:
                steps_data = stacked([c.id() for c in clients], local_steps,
                                     device=self.device)
            else:
                per_client = {c.id(): dataset.get_train_data(c.id(), local_steps)
                              for c in clients}
                steps_data = [self._stack_step(dataset, clients, s, per_client)
                              for s in range(local_steps)]

        # apply fused target transforms once per stacked step
        byz_tt_rows = [i for i, t in enumerate(target_tfms) if t is not None]

        def fix_targets(Y: Tensor) -> Tensor:
            if not byz_tt_rows:
                return Y
            Y = Y.clone()
            for i in byz_tt_rows:
                Y[i] = target_tfms[i](Y[i])
            return Y

        fedsgd = local_steps == 1
        if fedsgd:
            grad_fn = self._build_grad_fn(shared_params=True)
            params = tuple(t for _, t in self.spec.named_slices(theta))
            X, Y = steps_data[0]
            Y = fix_targets(Y)
            grads = grad_fn(params, X, Y, clamp_hi, grad_sign)
            U = out if out is not None else torch.empty(
                (C, self.spec.d), device=self.device)
            views = self.spec.batched_views(U)
            neg_lr = -lr if not isinstance(lr, torch.Tensor) else lr.neg()
            with torch.no_grad():
                # fused scale+store: one slab pass instead of copy + mul_
                for name, g in zip(self.spec.names, grads):
                    torch.mul(g, neg_lr, out=views[name])
            return U

        # general FedAvg path: per-client divergent weights in a [C, d]
        # slab.  The buffer persists across rounds (the per-round
        # `theta.repeat` allocation was ~25% of the config-3 step,
        # VERDICT r1 weak #6); only the broadcast-copy of θ remains.
        if (getattr(self, "_slab", None) is None
                or self._slab.shape[0] != C):
            self._slab = torch.empty(C, self.spec.d, device=self.device)
        slab = self._slab
        slab.copy_(theta.unsqueeze(0).expand(C, -1))
        slab_views = self.spec.batched_views(slab)
        grad_fn = self._build_grad_fn(shared_params=False)
        for s in range(local_steps):
            X, Y = steps_data[s]
            Y = fix_targets(Y)
            params = tuple(slab_views[n] for n in self.spec.names)
            grads = grad_fn(params, X, Y, clamp_hi, grad_sign)
            with torch.no_grad():
                views = [slab_views[n] for n in self.spec.names]
                if isinstance(lr, torch.Tensor):
                    gl = list(grads)
                    torch._foreach_mul_(gl, -lr)
                    torch._foreach_add_(views, gl)
                else:
                    torch._foreach_add_(views, list(grads), alpha=-lr)
        slab.sub_(theta.unsqueeze(0))
        if out is not None:
            out.copy_(slab)
            return out
        return slab

    # ------------------------------------------------------------------ eval
    @torch.no_grad()
    def evaluate(self, theta: Tensor, clients: List[BladesClient], dataset,
                 round_number: int, batch_size: int, metrics) -> List[dict]:
        """Evaluate the GLOBAL model on every client's test shard.

        The reference pushes the same global weights to each client and
        evaluates per client (actor.py:35-48); since the weights are
        identical, this runs as plain batched inference per client shard.
        """
        params = dict(self.spec.named_slices(theta))
        self.base.eval()
        stacked = getattr(dataset, "get_stacked_test_data", None)
        if stacked is not None and clients:
            try:
                out = self._evaluate_stacked(params, stacked, clients,
                                             round_number, batch_size, metrics)
                self.base.train()
                return out
            except NotImplementedError:
                pass
        results = []
        for c in clients:
            test_set = dataset.get_all_test_data(c.id())
            loader = torch.utils.data.DataLoader(test_set, batch_size=batch_size)
            r = {"_meta": {"type": "client_validation"}, "E": round_number,
                 "Client": c.id(), "Length": 0, "Loss": 0.0}
            for name in metrics:
                r[name] = 0.0
            for data, target in loader:
                data = data.to(self.device)
                target = target.to(self.device)
                out = functional_call(self.base, (params, self.buffers), (data,))
                r["Loss"] += nn.functional.cross_entropy(out, target).item() * len(target)
                r["Length"] += len(target)
                for name, metric in metrics.items():
                    r[name] += metric(out, target) * len(target)
            for name in metrics:
                r[name] /= max(r["Length"], 1)
            r["Loss"] /= max(r["Length"], 1)
            results.append(r)
        self.base.train()
        return results

    @torch.no_grad()
    def _evaluate_stacked(self, params, stacked_fn, clients, round_number,
                          batch_size, metrics):
        """Batched eval: all shard clients' test sets in a few big forwards
        (weights are identical across clients — the per-client loop only
        cost host time)."""
        X, Y = stacked_fn([c.id() for c in clients], device=self.device)
        C, n = X.shape[0], X.shape[1]
        # BN in batch-stats mode (make_vmap_safe strips running stats) would
        # compute eval statistics across whichever clients share a chunk —
        # logits would depend on chunk size and neighbours.  Evaluate one
        # client per forward in that case so stats stay within a client's
        # own shard (deterministic, chunk-independent).
        has_batch_stats_bn = any(
            isinstance(m, nn.modules.batchnorm._BatchNorm)
            and m.running_mean is None
            for m in self.base.modules())
        if has_batch_stats_bn:
            chunk = 1
        else:
            chunk = max(1, (batch_size * 64) // max(n, 1))
        outs = []
        for i in range(0, C, chunk):
            xb = X[i:i + chunk].flatten(0, 1)
            o = functional_call(self.base, (params, self.buffers), (xb,))
            outs.append(o.view(-1, n, o.shape[-1]))
        O = torch.cat(outs)  # [C, n, num_classes]
        losses = nn.functional.cross_entropy(
            O.reshape(C * n, -1), Y.reshape(C * n), reduction="none"
        ).view(C, n).mean(1)
        results = []
        for i, c in enumerate(clients):
            r = {"_meta": {"type": "client_validation"}, "E": round_number,
                 "Client": c.id(), "Length": n,
                 "Loss": float(losses[i].item())}
            for name, metric in metrics.items():
                r[name] = metric(O[i], Y[i])
            results.append(r)
        return results
