"""Config-driven DAG network builder — MI355X-native replacement for the
reference's absent ``baseline.baseAgent`` submodule.

The reference consumes this API everywhere (call sites cited in SURVEY.md §2.8:
e.g. /root/reference/APE_X/Player.py:108, /root/reference/APE_X/Learner.py:127,
/root/reference/R2D2/Player.py:103-113, /root/reference/IMPALA/Learner.py:72).
The cfg ``model`` section is a node map: ``netCat`` (layer kind), ``prior``
(topological stage), ``prevNodeNames`` (DAG edges), ``input`` (graph input
indices), ``output`` (graph output flag) — see cfg/ape_x.json:37-88.

Supported netCat kinds (the 7 the reference's configs use):
  CNN2D, MLP, LSTMNET, ViewV2, Add, Mean, Substract
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn

_ACTS = {
    "relu": nn.ReLU,
    "leakyrelu": nn.LeakyReLU,
    "tanh": nn.Tanh,
    "sigmoid": nn.Sigmoid,
    "elu": nn.ELU,
    "linear": nn.Identity,
    "identity": nn.Identity,
}


def _act(name: str) -> nn.Module:
    return _ACTS[name.lower()]()


class CNN2D(nn.Module):
    """Conv stack per the reference cfg convention: ``nLayer`` counts layers
    including the trailing flatten when ``linear``/``fSize[-1]==-1``
    (cfg/ape_x.json:38-51: 4 layers = 3 convs + flatten)."""

    def __init__(self, cfg: Dict[str, Any]):
        super().__init__()
        in_ch = int(cfg["iSize"])
        n_layer = int(cfg["nLayer"])
        f_size = list(cfg["fSize"])
        n_unit = list(cfg["nUnit"])
        stride = list(cfg.get("stride", [1] * len(n_unit)))
        padding = list(cfg.get("padding", [0] * len(n_unit)))
        acts = list(cfg.get("act", ["relu"] * len(n_unit)))
        bn = list(cfg.get("BN", [False] * n_layer))
        self.flatten = bool(cfg.get("linear", False)) or (f_size and f_size[-1] == -1)

        layers: List[nn.Module] = []
        n_conv = len(n_unit)
        ch = in_ch
        self._convs: List[nn.Conv2d] = []
        for i in range(n_conv):
            conv = nn.Conv2d(ch, n_unit[i], kernel_size=f_size[i],
                             stride=stride[i], padding=padding[i])
            self._convs.append(conv)
            layers.append(conv)
            if i < len(bn) and bn[i]:
                layers.append(nn.BatchNorm2d(n_unit[i]))
            layers.append(_act(acts[i]))
            ch = n_unit[i]
        self.body = nn.Sequential(*layers)
        # fused-path eligibility (static part): plain strided ReLU convs
        self._fusable_static = (
            all(a.lower() == "relu" for a in acts[:n_conv])
            and not any(bn[:n_conv])
            and all(p == 0 for p in padding[:n_conv])
        )
        self._fused_checked: dict = {}

    def _fused_ok(self, x: torch.Tensor) -> bool:
        """Use the hand-written MFMA conv kernels (ops/hip/conv_mfma.hip)
        when the whole stack's geometry is covered and weights are bf16
        channels_last on a GPU."""
        if not (self._fusable_static and x.is_cuda):
            return False
        if x.dtype not in (torch.uint8, torch.bfloat16):
            return False
        key = (tuple(x.shape[1:]), x.dtype)
        hit = self._fused_checked.get(key)
        if hit is not None:
            return hit
        from .. import ops as _ops

        ok = self._convs[0].weight.dtype == torch.bfloat16 and x.is_contiguous(
            memory_format=torch.channels_last
        )
        if ok:
            C, H, W = x.shape[1:]
            u8 = x.dtype == torch.uint8
            for conv in self._convs:
                COUT, _, KH, KW = conv.weight.shape
                S = conv.stride[0]
                if not _ops.conv_supported(H, W, C, KH, KW, S, COUT, u8):
                    ok = False
                    break
                H = (H - KH) // S + 1
                W = (W - KW) // S + 1
                C = COUT
                u8 = False
        self._fused_checked[key] = ok
        return ok

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._fused_ok(x):
            from .. import ops as _ops

            y = x
            last = len(self._convs) - 1
            for i, conv in enumerate(self._convs):
                # last conv of a flattening stack writes the 2-D CHW-ordered
                # output directly (flatten fused into the epilogue)
                if i == last and self.flatten and _ops.conv_chw_supported(
                        y.shape[2], y.shape[3], y.shape[1],
                        conv.weight.shape[2], conv.weight.shape[3],
                        conv.stride[0], conv.weight.shape[0]):
                    return _ops.fused_conv_relu(y, conv.weight, conv.bias,
                                                conv.stride[0], chw_out=True)
                y = _ops.fused_conv_relu(y, conv.weight, conv.bias,
                                         conv.stride[0])
        else:
            y = self.body(x)
        if self.flatten:
            y = torch.flatten(y, 1)
        return y


class MLP(nn.Module):
    def __init__(self, cfg: Dict[str, Any]):
        super().__init__()
        in_f = int(cfg["iSize"])
        f_size = list(cfg["fSize"])
        acts = list(cfg.get("act", ["relu"] * len(f_size)))
        bn = list(cfg.get("BN", [False] * len(f_size)))
        layers: List[nn.Module] = []
        f = in_f
        for i, out_f in enumerate(f_size):
            layers.append(nn.Linear(f, out_f))
            if i < len(bn) and bn[i]:
                layers.append(nn.BatchNorm1d(out_f))
            layers.append(_act(acts[i]))
            f = out_f
        self.body = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.body(x)


class ViewV2(nn.Module):
    """Reshape node (cfg/r2d2.json:50-54): reshapes the previous node's output
    by a shape-hint tensor passed as a graph input (e.g. [seq, batch, -1];
    R2D2/Player.py:112-113, R2D2/Learner.py:97,107)."""

    def forward(self, x: torch.Tensor, shape_hint: torch.Tensor) -> torch.Tensor:
        shape = [int(v) for v in shape_hint.flatten().tolist()]
        return x.reshape(shape)


class LSTMNET(nn.Module):
    """Single nn.LSTM with externally managed cell state.

    ``FlattenMode`` (cfg/r2d2.json:55-65): input arrives as (seq, batch, feat)
    (produced by the preceding ViewV2), output is flattened back to
    (seq*batch, hidden) so per-step heads (MLP) apply uniformly.
    Cell state persists across forward calls until zeroed/overwritten —
    matching getCellState/setCellState/zeroCellState/detachCellState usage at
    R2D2/Player.py:103,149,173,213,260-261 and R2D2/Learner.py:86-87,103-104.
    """

    def __init__(self, cfg: Dict[str, Any]):
        super().__init__()
        self.hidden_size = int(cfg["hiddenSize"])
        self.input_size = int(cfg["iSize"])
        self.num_layers = int(cfg.get("nLayer", 1))
        self.flatten_mode = bool(cfg.get("FlattenMode", True))
        self.return_hidden = bool(cfg.get("return_hidden", False))
        self.lstm = nn.LSTM(self.input_size, self.hidden_size, self.num_layers)
        self._state: Optional[Tuple[torch.Tensor, torch.Tensor]] = None

    # -- cell-state management ------------------------------------------
    def get_cell_state(self):
        return self._state

    def set_cell_state(self, state) -> None:
        self._state = state

    def zero_cell_state(self, batch: int = 1) -> None:
        p = next(self.lstm.parameters())
        z = torch.zeros(
            self.num_layers, batch, self.hidden_size, device=p.device, dtype=p.dtype
        )
        self._state = (z, z.clone())

    def detach_cell_state(self) -> None:
        if self._state is not None:
            self._state = (self._state[0].detach(), self._state[1].detach())

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        p = next(self.lstm.parameters())
        # bf16 input projection (default ON since round 2; DRL_LSTM_BF16_IN=0
        # opts out): keep bf16 trunk features bf16 and run the big input
        # projection GEMMs in bf16 inside manual_lstm_seq — the recurrence,
        # gates and cell state stay fp32. GPU-numerics-validated against the
        # fp32 path (tests/test_gpu_algos.py) and measured 12% faster at
        # 32x80 (gpurun_out/r2_r2d2_bf16.json: 5.07 vs 5.77 ms/step).
        import os as _os

        bf16_in = (
            x.dtype == torch.bfloat16 and x.is_cuda and self.num_layers == 1
            and _os.environ.get("DRL_LSTM_BF16_IN", "1") == "1"
        )
        if x.dtype != p.dtype and not bf16_in:
            # mixed-trunk path: bf16 conv features into the fp32 LSTM
            x = x.to(p.dtype)
        if x.dim() == 2:
            x = x.unsqueeze(0)  # (1, batch, feat)
        seq, batch, _ = x.shape
        if self._state is None or self._state[0].shape[1] != batch:
            self.zero_cell_state(batch)
        st = self._state
        if st[0].dtype != p.dtype or st[0].device != x.device:
            st = (st[0].to(x.device, p.dtype), st[1].to(x.device, p.dtype))
        if x.is_cuda and self.num_layers == 1:
            # K5-lite: fused-cell manual recurrence (hipGraph-capturable;
            # MIOpen's RNN path is not) — models/manual_lstm.py
            from .manual_lstm import manual_lstm_seq

            out, new_state = manual_lstm_seq(x, st, self.lstm)
        else:
            out, new_state = self.lstm(x, st)
        self._state = new_state
        if self.flatten_mode:
            out = out.reshape(seq * batch, self.hidden_size)
        return out


class ResidualBlock(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.c1 = nn.Conv2d(ch, ch, 3, padding=1)
        self.c2 = nn.Conv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        y = self.c1(torch.relu(x))
        y = self.c2(torch.relu(y))
        return x + y


class ImpalaResNet(nn.Module):
    """IMPALA deep residual torso (Espeholt et al. 2018, fig. 3): per section
    conv3x3 -> maxpool3x3 s2 -> nBlocks residual blocks; final ReLU+flatten.
    Used by cfg/impala_resnet.json (BASELINE.json config 3: 'IMPALA deep
    ResNet V-trace, 8x MI355X learner-DP'). Beyond the reference repo's
    capability (it only ships the shallow torso) — netCat: IMPALA_RESNET."""

    def __init__(self, cfg: Dict[str, Any]):
        super().__init__()
        in_ch = int(cfg.get("iSize", 4))
        channels = list(cfg.get("channels", [16, 32, 32]))
        n_blocks = int(cfg.get("nBlocks", 2))
        sections: List[nn.Module] = []
        ch = in_ch
        for out_ch in channels:
            sections.append(nn.Conv2d(ch, out_ch, 3, padding=1))
            sections.append(nn.MaxPool2d(3, stride=2, padding=1))
            for _ in range(n_blocks):
                sections.append(ResidualBlock(out_ch))
            ch = out_ch
        self.body = nn.Sequential(*sections)

    def forward(self, x):
        y = self.body(x)
        y = torch.relu(y)
        return torch.flatten(y, 1)


class _Arith(nn.Module):
    KIND = "add"

    def forward(self, *xs: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError


class Add(_Arith):
    def forward(self, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        return a + b


class Substract(_Arith):  # (sic) reference spelling, cfg/ape_x.json:83
    def forward(self, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        return a - b


class Mean(_Arith):
    def forward(self, a: torch.Tensor) -> torch.Tensor:
        return a.mean(dim=-1, keepdim=True)


_NETCAT = {
    "CNN2D": CNN2D,
    "MLP": MLP,
    "LSTMNET": LSTMNET,
    "VIEWV2": ViewV2,
    "ADD": Add,
    "MEAN": Mean,
    "SUBSTRACT": Substract,
    "IMPALA_RESNET": ImpalaResNet,
}


class BaseAgent(nn.Module):
    """DAG network built from a cfg ``model`` dict.

    Execution order: ascending ``prior``, then node-name order within a stage.
    A node's inputs are [outputs of ``prevNodeNames``...] + [forward_args[i]
    for i in ``input``] (ViewV2 consumes its shape hint this way).
    ``forward`` takes and returns *lists* of tensors, matching the reference's
    ``baseAgent`` call sites.
    """

    def __init__(self, model_cfg: Dict[str, Any]):
        super().__init__()
        self.cfg = dict(model_cfg)
        self.node_order: List[str] = sorted(
            self.cfg.keys(), key=lambda k: (int(self.cfg[k].get("prior", 0)), k)
        )
        self.nodes = nn.ModuleDict()
        self.output_nodes: List[str] = []
        for name in self.node_order:
            ncfg = self.cfg[name]
            kind = str(ncfg["netCat"]).upper()
            if kind not in _NETCAT:
                raise ValueError(f"unsupported netCat {ncfg['netCat']!r} for node {name}")
            needs_cfg = kind in ("CNN2D", "MLP", "LSTMNET", "IMPALA_RESNET")
            self.nodes[name] = _NETCAT[kind](ncfg) if needs_cfg else _NETCAT[kind]()
            if ncfg.get("output", False):
                self.output_nodes.append(name)
        if not self.output_nodes:
            # degenerate cfg: last node is the output
            self.output_nodes = [self.node_order[-1]]
        self._lstm_nodes = [
            n for n in self.node_order if isinstance(self.nodes[n], LSTMNET)
        ]
        # detect the dueling pattern Substract(Add(A,V), Mean(A)) so it can
        # run as ONE fused kernel on GPU (ops K3) instead of three graph nodes
        consumers: Dict[str, int] = {}
        for name in self.node_order:
            for p in self.cfg[name].get("prevNodeNames", []):
                consumers[p] = consumers.get(p, 0) + 1
        self._dueling: Dict[str, Tuple[str, str]] = {}
        self._dueling_skip: set = set()
        for name in self.node_order:
            ncfg = self.cfg[name]
            if str(ncfg.get("netCat", "")).upper() != "SUBSTRACT":
                continue
            prevs = ncfg.get("prevNodeNames", [])
            if len(prevs) != 2:
                continue
            addn, meann = prevs
            acfg = self.cfg.get(addn, {})
            mcfg = self.cfg.get(meann, {})
            if (str(acfg.get("netCat", "")).upper() == "ADD"
                    and str(mcfg.get("netCat", "")).upper() == "MEAN"):
                aprev = acfg.get("prevNodeNames", [])
                mprev = mcfg.get("prevNodeNames", [])
                if (len(aprev) == 2 and mprev == [aprev[0]]
                        and consumers.get(addn) == 1
                        and consumers.get(meann) == 1):
                    self._dueling[name] = (aprev[0], aprev[1])
                    self._dueling_skip.update((addn, meann))

    # ---- forward -------------------------------------------------------
    def forward(self, inputs: Sequence[torch.Tensor]) -> List[torch.Tensor]:
        if isinstance(inputs, torch.Tensor):
            inputs = [inputs]
        produced: Dict[str, torch.Tensor] = {}
        use_fused_dueling = bool(self._dueling) and inputs[0].is_cuda
        for name in self.node_order:
            if use_fused_dueling:
                if name in self._dueling_skip:
                    continue
                if name in self._dueling:
                    from .. import ops as _ops

                    a_name, v_name = self._dueling[name]
                    produced[name] = _ops.dueling_head(
                        produced[a_name].float(), produced[v_name].float()
                    )
                    continue
            ncfg = self.cfg[name]
            args: List[torch.Tensor] = [
                produced[p] for p in ncfg.get("prevNodeNames", [])
            ]
            for idx in ncfg.get("input", []):
                args.append(inputs[int(idx)])
            produced[name] = self.nodes[name](*args)
        return [produced[n] for n in self.output_nodes]

    # ---- parameter utilities (reference baseAgent surface) -------------
    def getParameters(self) -> List[nn.Parameter]:
        return list(self.parameters())

    @torch.no_grad()
    def updateParameter(self, src: "BaseAgent", tau: float) -> None:
        """Polyak/hard update: self = tau*src + (1-tau)*self
        (APE_X/Learner.py:204-208 uses tau=1 for the hard target sync)."""
        if tau >= 1.0:
            self.load_state_dict(src.state_dict())
            return
        for p_t, p_s in zip(self.parameters(), src.parameters()):
            p_t.mul_(1.0 - tau).add_(p_s.to(p_t.device), alpha=tau)
        for b_t, b_s in zip(self.buffers(), src.buffers()):
            b_t.copy_(b_s.to(b_t.device))

    def calculateNorm(self) -> torch.Tensor:
        total = None
        for p in self.parameters():
            if p.grad is not None:
                n = p.grad.detach().float().norm(2) ** 2
                total = n if total is None else total + n
        if total is None:
            return torch.tensor(0.0)
        return total.sqrt()

    def clippingNorm(self, max_norm: float) -> torch.Tensor:
        return torch.nn.utils.clip_grad_norm_(self.parameters(), max_norm)

    # ---- LSTM cell-state surface ---------------------------------------
    def _lstm(self) -> LSTMNET:
        if not self._lstm_nodes:
            raise RuntimeError("model has no LSTMNET node")
        return self.nodes[self._lstm_nodes[0]]

    def getCellState(self):
        return self._lstm().get_cell_state()

    def setCellState(self, state) -> None:
        self._lstm().set_cell_state(state)

    def zeroCellState(self, batch: int = 1) -> None:
        self._lstm().zero_cell_state(batch)

    def detachCellState(self) -> None:
        self._lstm().detach_cell_state()

    @property
    def has_lstm(self) -> bool:
        return bool(self._lstm_nodes)


# Lower-case alias matching the reference's class name (`baseAgent(MODEL)`)
baseAgent = BaseAgent


def get_optim(optim_info: Dict[str, Any], model: nn.Module) -> torch.optim.Optimizer:
    """Optimizer factory — reference ``baseline.utils.getOptim`` equivalent
    (cfg schemas: cfg/ape_x.json:27-35 rmsprop, cfg/r2d2.json:28-32 adam)."""
    info = dict(optim_info)
    name = str(info.pop("name")).lower()
    params = model.parameters()
    if name == "rmsprop":
        return torch.optim.RMSprop(
            params,
            lr=float(info.get("lr", 1e-3)),
            eps=float(info.get("eps", 1e-8)),
            weight_decay=float(info.get("decay", 0.0)),
            alpha=float(info.get("alpha", 0.99)),
            momentum=float(info.get("momentum", 0.0)),
            centered=bool(info.get("centered", False)),
            foreach=True,
        )
    if name == "adam":
        return torch.optim.Adam(
            params,
            lr=float(info.get("lr", 1e-3)),
            eps=float(info.get("eps", 1e-8)),
            weight_decay=float(info.get("decay", 0.0)),
            foreach=True,
        )
    if name == "sgd":
        return torch.optim.SGD(
            params,
            lr=float(info.get("lr", 1e-2)),
            momentum=float(info.get("momentum", 0.0)),
            weight_decay=float(info.get("decay", 0.0)),
            foreach=True,
        )
    raise ValueError(f"unknown optimizer {name!r}")


getOptim = get_optim
