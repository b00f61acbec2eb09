"""Deeper engine behaviors mirroring the reference's per-family test files:
label-flip against a real model, Byzantine nodes inside P2P gossip, payload
serialization round-trips, application pipeline-name reservations."""
import asyncio

import numpy as np
import pytest
import torch
import torch.nn as nn

from byzpy_amd.attacks import LabelFlipAttack


class TestLabelFlipModel:
    def _model_and_batch(self):
        torch.manual_seed(0)
        model = nn.Linear(6, 3)
        x = torch.randn(8, 6)
        y = torch.randint(0, 3, (8,))
        return model, (x, y)

    def test_mirror_matches_manual(self):
        model, (x, y) = self._model_and_batch()
        atk = LabelFlipAttack(num_classes=3)
        out = atk.apply(model=model, batch=(x, y))
        model.zero_grad(set_to_none=True)
        loss = nn.CrossEntropyLoss()(model(x), 2 - y)
        ref = torch.autograd.grad(loss, list(model.parameters()))
        ref_flat = torch.cat([g.reshape(-1) for g in ref])
        assert torch.allclose(out, ref_flat, atol=1e-6)

    def test_explicit_mapping(self):
        model, (x, y) = self._model_and_batch()
        mapping = {0: 1, 1: 0, 2: 2}
        out = atk_out = LabelFlipAttack(mapping=mapping).apply(model=model, batch=(x, y))
        flipped = y.clone()
        flipped[y == 0] = 1
        flipped[y == 1] = 0
        model.zero_grad(set_to_none=True)
        loss = nn.CrossEntropyLoss()(model(x), flipped)
        ref = torch.cat(
            [g.reshape(-1) for g in torch.autograd.grad(loss, list(model.parameters()))]
        )
        assert torch.allclose(out, ref, atol=1e-6)

    def test_scale(self):
        model, (x, y) = self._model_and_batch()
        a = LabelFlipAttack(num_classes=3, scale=1.0).apply(model=model, batch=(x, y))
        b = LabelFlipAttack(num_classes=3, scale=-2.0).apply(model=model, batch=(x, y))
        assert torch.allclose(b, -2.0 * a, atol=1e-5)


class TestP2PWithByzantine:
    def test_honest_nodes_resist_byzantine_peer(self):
        """3 honest (zero gradient => theta-half = theta) + 1 Byzantine
        (scaled empire) on a complete topology: every honest node must stay
        inside the honest parameter envelope after robust gossip rounds."""
        from byzpy_amd.aggregators import CoordinateWiseMedian
        from byzpy_amd.attacks import EmpireAttack
        from byzpy_amd.engine.peer_to_peer.mixin import (
            P2PByzantineMixin,
            P2PHonestMixin,
        )
        from byzpy_amd.engine.peer_to_peer.train import PeerToPeer

        d = 8

        class StaticHonest(P2PHonestMixin):
            def __init__(self, seed):
                self.model = nn.Linear(d, 1, bias=False)
                g = torch.Generator().manual_seed(seed)
                with torch.no_grad():
                    self.model.weight.copy_(torch.randn(1, d, generator=g))
                self.lr = 0.1

            def p2p_local_loss_backward(self):
                (0.0 * self.model.weight.sum()).backward()

        class Byz(P2PByzantineMixin):
            def __init__(self):
                self.attack = EmpireAttack(scale=-25.0)

        honest = [StaticHonest(s) for s in range(3)]
        starts = [h.p2p_flat_params().clone() for h in honest]

        async def main():
            p2p = PeerToPeer(honest, [Byz()], CoordinateWiseMedian())
            await p2p.bootstrap()
            try:
                for _ in range(3):
                    await p2p.round()
            finally:
                await p2p.shutdown()

        asyncio.run(main())
        lo = torch.stack(starts).min(dim=0).values - 1e-4
        hi = torch.stack(starts).max(dim=0).values + 1e-4
        for h in honest:
            v = h.p2p_flat_params()
            assert (v >= lo).all() and (v <= hi).all()


class TestPayloadSerialization:
    def test_wrap_unwrap_nested(self):
        from byzpy_amd.actor.ipc import unwrap_payload, wrap_payload

        payload = {
            "t": torch.arange(6.0),
            "nested": [np.ones(3, dtype=np.float32), {"k": torch.zeros(2, 2)}],
            "plain": 42,
        }
        wrapped = wrap_payload(payload)
        out = unwrap_payload(wrapped)
        assert torch.equal(out["t"], torch.arange(6.0))
        assert np.allclose(out["nested"][0], np.ones(3))
        assert torch.equal(out["nested"][1]["k"], torch.zeros(2, 2))
        assert out["plain"] == 42

    def test_bf16_tensor_roundtrip(self):
        from byzpy_amd.actor.ipc import unwrap_payload, wrap_payload

        t = torch.randn(17).bfloat16()
        out = unwrap_payload(wrap_payload({"g": t}))
        assert out["g"].dtype == torch.bfloat16
        assert torch.equal(out["g"], t)


class TestApplicationReservations:
    def test_honest_reserved_names(self):
        from byzpy_amd.engine.node.application import HonestNodeApplication
        from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
        from byzpy_amd.graph.ops import CallableOp

        app = HonestNodeApplication(pool=None)
        g = ComputationGraph(
            [GraphNode("x", CallableOp(lambda v: v), {"v": GraphInput("v")})]
        )
        with pytest.raises(ValueError):
            app.register_pipeline("aggregate", g)

    def test_byzantine_reserved_names(self):
        from byzpy_amd.engine.node.application import ByzantineNodeApplication
        from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
        from byzpy_amd.graph.ops import CallableOp

        app = ByzantineNodeApplication(pool=None)
        g = ComputationGraph(
            [GraphNode("x", CallableOp(lambda v: v), {"v": GraphInput("v")})]
        )
        with pytest.raises(ValueError):
            app.register_pipeline("attack", g)
