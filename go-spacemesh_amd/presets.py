"""POST config presets — the reference's preset registry restricted to the
POST surface (config/presets/presets.go:9-34; parameter sources cited per
preset).  `get(name)` returns (PostConfig, PostSetupOpts-defaults)."""
from __future__ import annotations

from typing import Dict, Tuple

from .api import PostConfig, PostSetupOpts

MAINNET_POW_DIFFICULTY = bytes.fromhex(
    "000dfb23b0979b4b000000000000000000000000000000000000000000000000")


def mainnet() -> Tuple[PostConfig, PostSetupOpts]:
    """config/mainnet.go:183-191 (+ scrypt N=8192 dep default,
    activation/post.go:155; 288 nonces config/mainnet.go:61)."""
    return (PostConfig(min_num_units=4, max_num_units=2**32 - 1,
                       labels_per_unit=4294967296, k1=26, k2=37, k3=1,
                       pow_difficulty=MAINNET_POW_DIFFICULTY),
            PostSetupOpts(num_units=4, scrypt_n=8192,
                          max_file_size=4294967296))


def testnet() -> Tuple[PostConfig, PostSetupOpts]:
    """config/presets/testnet.go:136-144."""
    return (PostConfig(min_num_units=2, max_num_units=2**32 - 1,
                       labels_per_unit=1024, k1=26, k2=37, k3=1,
                       pow_difficulty=MAINNET_POW_DIFFICULTY),
            PostSetupOpts(num_units=2, scrypt_n=8192))


def fastnet() -> Tuple[PostConfig, PostSetupOpts]:
    """config/presets/fastnet.go:68-81 (toy POST: K1=12 K2=4 K3=1,
    LabelsPerUnit=128, 2..4 units)."""
    return (PostConfig(min_num_units=2, max_num_units=4,
                       labels_per_unit=128, k1=12, k2=4, k3=1,
                       pow_difficulty=MAINNET_POW_DIFFICULTY),
            PostSetupOpts(num_units=2, scrypt_n=2))


_REGISTRY: Dict[str, object] = {"mainnet": mainnet, "testnet": testnet,
                                "fastnet": fastnet}


def get(name: str) -> Tuple[PostConfig, PostSetupOpts]:
    if name not in _REGISTRY:
        raise KeyError(f"unknown preset {name!r}; have "
                       f"{sorted(_REGISTRY)}")
    return _REGISTRY[name]()
