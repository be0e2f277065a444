import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (MI355X); run with -m gpu on a GPU box")


@pytest.fixture
def fake_embed():
    """Deterministic embedding fn for tests: hashed character trigrams."""
    import hashlib

    import numpy as np

    def embed(texts):
        out = []
        for t in texts:
            v = np.zeros(64, dtype=float)
            s = t.lower()
            for i in range(max(1, len(s) - 2)):
                gram = s[i : i + 3]
                h = int(hashlib.md5(gram.encode()).hexdigest(), 16)
                v[h % 64] += 1.0
            n = np.linalg.norm(v)
            out.append((v / n if n else v).tolist())
        return out

    return embed


# Deep-fuzz profile: KLLMS_FUZZ_DEEP=<mult> multiplies every test's
# max_examples (used for long validation campaigns; default off)
try:
    from hypothesis import settings as _hyp_settings

    _mult = int(os.environ.get("KLLMS_FUZZ_DEEP", "0"))
    if _mult > 1:
        _hyp_settings.register_profile(
            "deep", max_examples=_mult * 100, deadline=None, print_blob=True)
        _hyp_settings.load_profile("deep")
except Exception:
    pass
