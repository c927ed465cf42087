"""Cluster status model: live view of a nodes_map deployment.

Counterpart of the reference's `ControlCenter`
(/root/reference/distllm/control_center.py:8-71) — there an in-memory
model with stubbed operations; here a working aggregator that polls every
node of a cluster config over the control plane and validates a pushed
model's slice layout against the hparams, which is what the reference's
`push_model` checks locally (control_center.py:24-50).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Tuple

from .client import Connection, parse_address


@dataclass
class ModelSlice:
    name: str
    layer_from: int
    layer_to: int


@dataclass
class NodeStatus:
    connectivity: bool = False
    model: str = ""
    first_layer: int = -1
    n_layers: int = 0
    device: str = ""
    error: str = ""

    @property
    def slice_loaded(self) -> bool:
        return bool(self.model)


class ControlCenter:
    """Polls the nodes of a `nodes_map` and reports cluster health."""

    def __init__(self, nodes_map: Dict[str, List[int]]):
        self.nodes_map = dict(nodes_map)

    def validate_slices(self, n_layer: int,
                        slices: List[ModelSlice]) -> None:
        """The reference's push_model validation: the slices must tile
        [0, n_layer) contiguously (control_center.py:24-50 semantics)."""
        ranges = sorted((s.layer_from, s.layer_to) for s in slices)
        expect = 0
        for a, b in ranges:
            if a != expect or b < a:
                raise ValueError(
                    f"slices must tile layers contiguously; got {ranges}")
            expect = b + 1
        if expect != n_layer:
            raise ValueError(
                f"slices cover [0, {expect}) but the model has {n_layer} "
                "layers")

    def get_status(self) -> Dict[str, NodeStatus]:
        out: Dict[str, NodeStatus] = {}
        for addr in self.nodes_map:
            host, port = parse_address(addr)
            conn = Connection(host, port, timeout=5.0)
            st = NodeStatus()
            try:
                resp = conn.get_status()
                st.connectivity = True
                st.model = resp.model
                st.first_layer = resp.first_layer
                st.n_layers = resp.n_layers
                st.device = resp.device
            except Exception as e:  # noqa: BLE001
                st.error = f"{type(e).__name__}: {e}"
            finally:
                conn.close()
            out[addr] = st
        return out

    def pipeline_ready(self, n_layer: int) -> Tuple[bool, str]:
        """True when every node is up and the loaded slices tile the
        model's layers in nodes_map order."""
        status = self.get_status()
        covered = []
        for addr, (a, b) in sorted(self.nodes_map.items(),
                                   key=lambda kv: kv[1][0]):
            st = status[addr]
            if not st.connectivity:
                return False, f"{addr}: unreachable ({st.error})"
            if not st.slice_loaded:
                return False, f"{addr}: no slice loaded"
            if st.first_layer != a or st.n_layers != b - a + 1:
                return (False,
                        f"{addr}: holds layers [{st.first_layer}, "
                        f"{st.first_layer + st.n_layers - 1}], expected "
                        f"[{a}, {b}]")
            covered.append((a, b))
        try:
            self.validate_slices(
                n_layer, [ModelSlice("", a, b) for a, b in covered])
        except ValueError as e:
            return False, str(e)
        return True, "ok"
