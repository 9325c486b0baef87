"""`dtmx.symbol` — guidance stub. dtmx has NO symbolic graph layer by
design (SURVEY.md §1 layer 5 / PARITY.md "NNVM glue: n/a"): the substrate
is eager torch autograd, with hipGraph capture (`Module.graphed_step`)
covering the launch-overhead role of symbolic executors. This module exists
so `import dtmx.symbol` / `mx.sym.*` call sites fail with directions
instead of AttributeError."""
from __future__ import annotations

_MSG = (
    "dtmx has no symbolic graph API (mx.sym.{}). Build models imperatively:\n"
    "  - model zoo: dtmx.models.get_symbol(name, ...) -> nn.Module\n"
    "  - layers:    dtmx.ops.layers (Conv2dNHWC, BatchNorm2dNHWC, ...)\n"
    "  - training:  dtmx.Module(net).fit(...) — same surface as the\n"
    "    reference's module API; hipGraph capture via DTMX_HIPGRAPH=1\n"
    "    replaces symbolic-executor bulking."
)


def __getattr__(name: str):
    raise AttributeError(_MSG.format(name))


class Symbol:
    def __init__(self, *a, **k):
        raise TypeError(_MSG.format("Symbol"))
