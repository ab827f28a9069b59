"""Client layer: per-client state machine + Byzantine base class.

API parity with the reference client (reference: src/blades/client.py:12-253):
the full lifecycle-hook surface (``set_model/set_para/set_lr/set_loss``,
``on_train_round_begin/end``, ``on_train_batch_begin``, ``local_training``,
``evaluate``, ``get_update/save_update``) exists and behaves identically, so
user subclasses written against the reference run unchanged on the loop
engine.

MI355X-native difference: clients are *descriptors*, not workers.  The
fused many-model engine (blades_amd.engine) trains every client whose
training hooks are un-overridden in one batched pass on the GPU; a client
only materializes its own ``self.model`` (the reference deep-copied one per
client up front — client.py:88) when user code actually drives it through
the per-client hooks.

Update semantics (unchanged): update = θ_after − θ_before over the local
round, flattened over requires_grad params in named_parameters order
(reference: client.py:130,216-228), ``nan_to_num`` on read (client.py:198).
"""
from __future__ import annotations

import copy
import logging
from collections import defaultdict
from typing import Optional

import torch
import torch.nn as nn
from torch.utils.data import DataLoader


class BladesClient:
    _is_byzantine: bool = False
    _is_trusted: bool = False
    device: str = "cpu"

    def __init__(self, id: Optional[str] = None, device: Optional[str] = "cpu"):
        self._state = defaultdict(dict)
        self.set_id(id)
        self.device = device
        self._running = {}
        self._json_logger = logging.getLogger("stats")
        self.debug_logger = logging.getLogger("debug")
        self.model: Optional[nn.Module] = None
        self.optimizer = None
        self.loss_func = nn.CrossEntropyLoss()
        self._lr: float = 0.1

    # ------------------------------------------------------------- identity
    def set_id(self, id: str) -> None:
        """Sets the unique id of the client."""
        self._id = id

    def id(self) -> str:
        """Returns the unique id of the client.

        :Example:

        >>> from blades_amd.client import BladesClient
        >>> client = BladesClient(id='1')
        >>> client.id()
        '1'
        """
        return self._id

    def getattr(self, attr):
        return getattr(self, attr)

    def is_byzantine(self) -> bool:
        return self._is_byzantine

    def is_trusted(self) -> bool:
        return self._is_trusted

    def trust(self, trusted: Optional[bool] = True) -> None:
        self._is_trusted = trusted

    # ------------------------------------------------------------- training
    def set_model(self, model: nn.Module, opt: type = torch.optim.SGD,
                  lr: float = 0.1) -> None:
        """Deep-copy the given model to the client (reference: client.py:88-89).

        On the fused engine this copy is only made for clients that override
        the training hooks; standard clients train out of the shared slab.
        """
        self.model = copy.deepcopy(model)
        self.optimizer = opt(self.model.parameters(), lr=lr)
        self._lr = lr

    def set_lr(self, lr: float) -> None:
        self._lr = lr
        if self.optimizer is not None:
            for g in self.optimizer.param_groups:
                g["lr"] = lr

    def set_loss(self, loss_func: str = "crossentropy") -> None:
        if loss_func == "crossentropy":
            self.loss_func = nn.CrossEntropyLoss()
        else:
            raise NotImplementedError(loss_func)

    def set_para(self, model: nn.Module) -> None:
        """Load the global weights (reference: client.py:107-109)."""
        self.model.load_state_dict(model.state_dict())

    def on_train_round_begin(self, use_actor: bool = True) -> None:
        self._save_para()
        self.model = self.model.to(self.device)
        self.model.train()

    def on_train_round_end(self) -> None:
        update = self._get_para(current=True) - self._get_para(current=False)
        self.save_update(update)

    def on_train_batch_begin(self, data, target, logs=None):
        return data, target

    def local_training(self, data_batches: list) -> None:
        """Per-batch: zero_grad / forward / clamp(loss) / backward / step
        (reference: client.py:178-193; loss clamp [0, 1e6] at :191)."""
        for data, target in data_batches:
            data, target = data.to(self.device), target.to(self.device)
            data, target = self.on_train_batch_begin(data=data, target=target)
            self.optimizer.zero_grad()
            output = self.model(data)
            loss = torch.clamp(self.loss_func(output, target), 0, 1e6)
            loss.backward()
            self.optimizer.step()

    # ------------------------------------------------------------- updates
    def get_update(self) -> torch.Tensor:
        return torch.nan_to_num(self._get_saved_update())

    def save_update(self, update: torch.Tensor) -> None:
        self._state["saved_update"] = update.detach().clone()

    def save_update_view(self, view: torch.Tensor) -> None:
        """Engine-internal: store a row view of the rank-local update slab
        (zero-copy; attackers that overwrite via save_update replace it)."""
        self._state["saved_update"] = view

    def _get_saved_update(self) -> torch.Tensor:
        return self._state["saved_update"]

    def _save_para(self) -> None:
        for name, param in self.model.named_parameters():
            if not param.requires_grad:
                continue
            self._state["saved_para"][name] = param.data.detach().clone()

    def _get_para(self, current: bool = True) -> torch.Tensor:
        layer_parameters = []
        for name, param in self.model.named_parameters():
            if not param.requires_grad:
                continue
            if current:
                layer_parameters.append(param.data.view(-1))
            else:
                layer_parameters.append(self._state["saved_para"][name].view(-1))
        return torch.cat(layer_parameters).to("cpu")

    # ----------------------------------------------------------------- eval
    def evaluate(self, round_number, test_set, batch_size, metrics,
                 use_actor: bool = True) -> dict:
        dataloader = DataLoader(dataset=test_set, batch_size=batch_size)
        self.model.eval()
        r = {"_meta": {"type": "client_validation"}, "E": round_number,
             "Length": 0, "Loss": 0}
        for name in metrics:
            r[name] = 0
        with torch.no_grad():
            for data, target in dataloader:
                data, target = data.to(self.device), target.to(self.device)
                output = self.model(data)
                r["Loss"] += self.loss_func(output, target).item() * len(target)
                r["Length"] += len(target)
                for name, metric in metrics.items():
                    r[name] += metric(output, target) * len(target)
        for name in metrics:
            r[name] /= r["Length"]
        r["Loss"] /= r["Length"]
        return r

    def __str__(self) -> str:
        return "BladesClient"


class ByzantineClient(BladesClient):
    """Base class for Byzantine clients (reference: client.py:231-253).

    Override ``local_training`` / ``on_train_batch_begin`` /
    ``omniscient_callback`` to implement an attack — the same three hook
    points the reference exposes.
    """

    _is_byzantine = True

    def omniscient_callback(self, simulator) -> None:
        """Runs after every round's update gather with full system knowledge.
        Default: no-op."""
        pass


def uses_default_training(client: BladesClient) -> bool:
    """True when the client's training hooks are the stock ones, i.e. the
    fused many-model engine may batch it (engine-internal)."""
    lt = type(client).local_training
    hb = type(client).on_train_batch_begin
    rb = type(client).on_train_round_begin
    from blades_amd.attackers import FUSABLE_CLIENT_TYPES

    if type(client) in FUSABLE_CLIENT_TYPES:
        return True
    return (lt is BladesClient.local_training
            and hb is BladesClient.on_train_batch_begin
            and rb is BladesClient.on_train_round_begin)
