"""Metrics emission + preset registry (SURVEY §5 hooks)."""
import importlib

import pytest

import gsm_amd

metrics = importlib.import_module("go-spacemesh_amd.metrics")
presets = importlib.import_module("go-spacemesh_amd.presets")


def _sample(name, labels=None):
    return metrics.registry.get_sample_value(name, labels or {})


def test_presets_registry():
    cfg, opts = presets.get("mainnet")
    assert (cfg.labels_per_unit, cfg.k1, cfg.k2, cfg.k3) == \
        (4294967296, 26, 37, 1)
    assert cfg.pow_difficulty.hex().startswith("000dfb23b0979b4b")
    assert opts.scrypt_n == 8192 and opts.num_units == 4
    cfg, opts = presets.get("fastnet")
    assert (cfg.labels_per_unit, cfg.k1, cfg.k2, cfg.k3) == (128, 12, 4, 1)
    assert cfg.min_num_units == 2 and cfg.max_num_units == 4
    cfg, _ = presets.get("testnet")
    assert cfg.labels_per_unit == 1024 and cfg.min_num_units == 2
    with pytest.raises(KeyError):
        presets.get("nope")


def test_verifier_pool_emits_metrics():
    class Inner:
        def verify(self, proof, meta, opts):
            pass

    class Meta:
        node_id = b"\x01" * 32

    before = _sample("activation_post_verification_seconds_count") or 0
    pool = gsm_amd.OffloadingVerifier(Inner(), workers=2)
    for _ in range(5):
        pool.verify("p", Meta(), None)
    pool.close()
    after = _sample("activation_post_verification_seconds_count")
    assert after == before + 5
    assert _sample("activation_post_verification_waiting_total") == 0


def test_prove_emits_post_seconds(monkeypatch):
    # patch the inner prover so no GPU is needed
    import importlib
    api = importlib.import_module("go-spacemesh_amd.api")
    monkeypatch.setattr(api, "_prove_buffer",
                        lambda *a, **k: gsm_amd.PostProof(0, b"", 0))
    gsm_amd.api.prove_buffer(b"", 0, b"\x00" * 32, b"\x00" * 32,
                             bytes(32), gsm_amd.PostConfig(),
                             gsm_amd.ProveOpts())
    assert _sample("smh_post_seconds") is not None
    assert _sample("activation_post_duration") is not None
