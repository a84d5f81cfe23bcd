

def test_transitions_npz_roundtrip(tmp_path):
    import numpy as np
    from agilerl_amd.training import save_transitions, load_transitions

    ds = {
        "observations": np.random.randn(32, 4).astype(np.float32),
        "actions": np.random.randint(0, 2, 32),
        "rewards": np.random.randn(32).astype(np.float32),
        "next_observations": np.random.randn(32, 4).astype(np.float32),
        "terminals": np.zeros(32, dtype=np.float32),
    }
    p = str(tmp_path / "ds.npz")
    save_transitions(ds, p)
    back = load_transitions(p)
    assert set(back) == set(ds)
    np.testing.assert_array_equal(back["observations"], ds["observations"])


def test_capability_flags():
    import agilerl_amd as pkg

    # replaced stacks are reported absent; first-party paths are the API
    assert pkg.HAS_VLLM is False
    assert pkg.HAS_DEEPSPEED is False
    assert pkg.HAS_LIGER_KERNEL is False
    assert isinstance(pkg.HAS_LLM_DEPENDENCIES, bool)


def test_collect_transitions_random_policy():
    import numpy as np

    from agilerl_amd.envs import CartPoleVecEnv
    from agilerl_amd.training import collect_transitions, load_transitions_into_buffer
    from agilerl_amd.components import ReplayBuffer

    env = CartPoleVecEnv(num_envs=4, seed=0)
    ds = collect_transitions(env, steps=25)
    assert ds["observations"].shape == (100, 4)
    assert set(ds) == {"observations", "actions", "rewards",
                       "next_observations", "terminals"}
    buf = ReplayBuffer(200)
    load_transitions_into_buffer(ds, buf)
    assert len(buf) == 100
