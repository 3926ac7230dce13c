"""Human-readable identity mnemonics (reference ``cdn-proto/src/util.rs:13-23``):
a stable 64-bit hash of a byte identity rendered as two dictionary words."""

from __future__ import annotations

import hashlib

_WORDS = [
    "able", "acid", "aged", "also", "apex", "aqua", "arch", "atom",
    "aunt", "away", "axis", "back", "bald", "barn", "bead", "beam",
    "bell", "bird", "blue", "bold", "bone", "book", "boss", "brag",
    "brim", "bulk", "buzz", "cake", "calm", "cape", "card", "cave",
    "chef", "chip", "city", "clay", "club", "coal", "coat", "coil",
    "cold", "cone", "cork", "crab", "crew", "crop", "cube", "cure",
    "dark", "dawn", "dean", "deep", "dice", "dime", "dish", "dock",
    "dome", "door", "dove", "drum", "dune", "dusk", "each", "earl",
    "east", "echo", "edge", "envy", "epic", "even", "exit", "face",
    "fact", "fang", "farm", "fern", "fig", "film", "fire", "fish",
    "flag", "flat", "flux", "foam", "fog", "fork", "fort", "fox",
    "free", "frog", "fuel", "fund", "gate", "gear", "gem", "gift",
    "glow", "goat", "gold", "golf", "gray", "grid", "grip", "gulf",
    "hail", "half", "hall", "hand", "harp", "hawk", "haze", "heat",
    "herb", "hero", "hill", "hint", "hive", "holy", "home", "hoof",
    "hook", "horn", "host", "hour", "husk", "icon", "inch", "iris",
    "iron", "isle", "ivy", "jade", "jazz", "jeep", "join", "joke",
    "jolt", "july", "jump", "june", "jury", "kale", "keel", "keen",
    "kelp", "kick", "kind", "king", "kite", "knee", "knot", "lace",
    "lake", "lamb", "lamp", "land", "lark", "lava", "leaf", "lens",
    "lily", "lime", "lion", "loaf", "lock", "loft", "logo", "loop",
    "luck", "lung", "lute", "mail", "main", "malt", "mane", "map",
    "mare", "mask", "mast", "mate", "maze", "mead", "mesa", "mild",
    "milk", "mill", "mint", "mist", "mole", "moon", "moss", "moth",
    "myth", "nail", "name", "navy", "neat", "nest", "news", "node",
    "noon", "nose", "note", "nova", "oak", "oath", "obey", "odds",
    "only", "onyx", "opal", "open", "oval", "oven", "palm", "park",
    "peak", "pear", "peat", "pelt", "pier", "pike", "pine", "pint",
    "plum", "pond", "pony", "pool", "port", "post", "prow", "puma",
    "pure", "quay", "quiz", "rail", "rain", "ramp", "rare", "reef",
    "rice", "ride", "ring", "risk", "road", "rock", "root", "rose",
    "ruby", "rust", "sage", "sail", "salt", "sand", "seal", "seed",
]


def hash64(identity: bytes) -> int:
    """Stable 64-bit hash of a byte identity (reference util.rs:19-23)."""
    return int.from_bytes(hashlib.blake2b(identity, digest_size=8).digest(), "little")


def mnemonic(identity: bytes) -> str:
    h = hash64(identity)
    n = len(_WORDS)
    return f"{_WORDS[h % n]}-{_WORDS[(h >> 8) % n]}-{(h >> 16) & 0xFFFF:04x}"
