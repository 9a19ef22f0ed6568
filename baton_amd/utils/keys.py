"""Credential / identifier generation.

The reference derives keys from ``random.sample(ascii_letters, n)``
(/root/reference/utils.py:38-39) — non-cryptographic, capped at 52 chars and
never repeating a character (defect D6 in SURVEY.md §2.5). Here identifiers
and keys come from the ``secrets`` module.
"""

import secrets


def new_key(nbytes: int = 16) -> str:
    """Cryptographically random hex key (default 32 chars, like the
    reference's 32-char key at client_manager.py:94)."""
    return secrets.token_hex(nbytes)


def new_client_id(experiment: str, nchars: int = 6) -> str:
    """Client id with the reference's shape ``client_{exp}_{6char}``
    (client_manager.py:89-93) but a crypto-random suffix."""
    return f"client_{experiment}_{secrets.token_hex((nchars + 1) // 2)[:nchars]}"
