"""In-process end-to-end round trip on a real MI355X — the shape of the
reference's activation/e2e suite (nipost_test.go:151-231: real
PostSetupManager + supervisor + post-service + verifier asserting
init -> prove -> validate), plus the adversarial invalid-index /
malfeasance flow (systest distributed_post_verification_test.go:253-267,
malfeasance.go:161-169)."""
import importlib
import os

import pytest

import gsm_amd
from gsm_amd import events as ev
from oracle import Oracle

pytestmark = pytest.mark.gpu

sup_mod = importlib.import_module("go-spacemesh_amd.supervisor")

NODE = bytes([0xA5]) * 32
ATX = bytes([0x5A]) * 32
CHALLENGE = bytes([0x09]) * 32
POW_DIFF = bytes([0x0F]) + bytes([0xFF]) * 31


def test_full_node_shaped_roundtrip(tmp_path):
    d = str(tmp_path)
    NU, LPU, N = 2, 1 << 10, 32
    seen_events = []
    unsub = ev.bus().subscribe(seen_events.append)
    try:
        # 1. init through the manager into a real postdata dir
        cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=LPU,
                                 k1=12, k2=8, k3=4,
                                 pow_difficulty=POW_DIFF)
        opts = gsm_amd.PostSetupOpts(data_dir=d, num_units=NU, scrypt_n=N,
                                     max_file_size=1 << 14,
                                     scratch_bytes=8 << 30)
        mgr = gsm_amd.PostSetupManager(NODE, ATX, cfg, opts)
        mgr.prepare_initializer()
        mgr.start_session()
        vrf = mgr.vrf_nonce()
        mgr.reset()
        assert any(isinstance(e, ev.InitStart) for e in seen_events)
        assert any(isinstance(e, ev.InitComplete) for e in seen_events)

        # 2. out-of-process prover: supervisor + service child over gRPC
        server = sup_mod.PostServiceServer()
        sup = sup_mod.PostSupervisor(server.address, d, nonces=16,
                                     k1=12, k2=8, pow_difficulty=POW_DIFF)
        sup.start()
        try:
            client = server.wait_for_client(timeout=30, poll_interval=0.2)
            info = client.info()
            assert info.node_id == NODE
            assert info.nonce is not None  # persisted VRF nonce
            sproof = client.proof(CHALLENGE, timeout=120)
        finally:
            sup.stop()
            server.stop()
        proof = gsm_amd.PostProof(sproof.nonce, sproof.indices, sproof.pow)

        # 3. the node's validation side: verifier pool + subset seeds
        meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
        inner = gsm_amd.PostVerifier(cfg, scrypt_n=N)
        pool = gsm_amd.OffloadingVerifier(inner, workers=2,
                                          prioritized_ids=[NODE])
        pool.verify(proof, meta,
                    gsm_amd.VerifyOpts(subset_seed=b"gossip-peer-1"))
        pool.verify(proof, meta,
                    gsm_amd.VerifyOpts(subset_seed=b"gossip-peer-2"),
                    prioritized=True)
        pool.close()

        # VRF nonce from init verifies (VerifyVRFNonce, validation.go:277)
        assert vrf is not None
        inner.verify_vrf_nonce(meta, vrf[0])

        # 4. adversarial: corrupted proof -> ErrInvalidIndex with position;
        #    peers re-check exactly that index (SelectedIndex)
        o = Oracle()
        bad = bytearray(proof.indices)
        bad[0] ^= 0x5A
        bp = gsm_amd.PostProof(proof.nonce, bytes(bad), proof.pow)
        full = gsm_amd.PostVerifier(
            gsm_amd.PostConfig(min_num_units=1, labels_per_unit=LPU, k1=12,
                               k2=8, k3=8, pow_difficulty=POW_DIFF),
            scrypt_n=N)
        with pytest.raises(gsm_amd.EngineError) as ei:
            full.verify(bp, meta)
        assert ei.value.code == gsm_amd.api.Status.INVALID_INDEX
        invalid_pos = int(str(ei.value).rsplit("position", 1)[1].strip())
        # the malfeasance re-check of exactly that index also fails ...
        with pytest.raises(gsm_amd.EngineError):
            full.verify(bp, meta,
                        gsm_amd.VerifyOpts(selected_index=invalid_pos))
        # ... while the original proof at the same position passes
        full.verify(proof, meta,
                    gsm_amd.VerifyOpts(selected_index=invalid_pos))
    finally:
        unsub()
