"""End-to-end learning test: A2C on pure-python CartPole must actually learn.

Mirrors the reference's test/integration/test_a2c.py (trains CartPole and
asserts the tail of episode returns clears a bar).
"""
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "examples"))


@pytest.mark.timeout(600)
def test_a2c_cartpole_learns(tmp_path):
    os.chdir(tmp_path)  # logs.tsv etc. go to tmp
    import a2c

    returns = a2c.train(total_steps=40000, address="127.0.0.1:0", log=False, seed=3)
    # CartPole starts at ~20 return with a random policy; after 40k steps the
    # recent-episode mean should be well clear of that.
    assert returns, "no episodes finished"
    mean_ret = sum(returns) / len(returns)
    assert mean_ret >= 80, f"did not learn: mean recent return {mean_ret:.1f}"
