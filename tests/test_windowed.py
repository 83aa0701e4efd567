"""Cyclic windowed buffer tests (reference test_cyclic_windowed_buffer.cc
sync/overlap semantics)."""
import numpy as np
import pytest

from trtlab_amd.core.windowed import CyclicWindowedBuffer


def test_windows_with_overlap():
    buf = CyclicWindowedBuffer(window_size=8, overlap=3)
    data = np.arange(32, dtype=np.float32)
    buf.push(data)
    wins = buf.pop_windows()
    # stride = 5: windows start at 0, 5, 10, ...
    assert len(wins) == 5
    for i, w in enumerate(wins):
        assert np.array_equal(w, np.arange(i * 5, i * 5 + 8))
    # consecutive windows share `overlap` samples
    assert np.array_equal(wins[0][-3:], wins[1][:3])


def test_incremental_push():
    buf = CyclicWindowedBuffer(window_size=4, overlap=1)
    total = 0
    for chunk in np.split(np.arange(20, dtype=np.float32), 10):
        total += buf.push(chunk)
    wins = buf.pop_windows()
    assert total == len(wins)
    assert np.array_equal(wins[0], [0, 1, 2, 3])
    assert np.array_equal(wins[1], [3, 4, 5, 6])


def test_callback_mode():
    seen = []
    buf = CyclicWindowedBuffer(window_size=4, overlap=0,
                               on_window=lambda w, i: seen.append((i, w.sum())))
    buf.push(np.ones(12, np.float32))
    assert [i for i, _ in seen] == [0, 1, 2]
    assert all(s == 4.0 for _, s in seen)


def test_multichannel_samples():
    buf = CyclicWindowedBuffer(window_size=4, overlap=2, sample_shape=(3,))
    buf.push(np.arange(24, dtype=np.float32).reshape(8, 3))
    wins = buf.pop_windows()
    assert len(wins) == 3
    assert wins[0].shape == (4, 3)
    assert np.array_equal(wins[0][2:], wins[1][:2])


@pytest.mark.gpu
def test_device_windowed_stack_roundtrip():
    from trtlab_amd import native
    from trtlab_amd.core.windowed import DeviceCyclicWindowedStack

    C = native()
    stack = DeviceCyclicWindowedStack(window_size=256, overlap=64,
                                      sample_bytes=4, capacity_windows=4)
    host = CyclicWindowedBuffer(window_size=256, overlap=64, dtype=np.float32)
    host.push(np.arange(1024, dtype=np.float32))
    for w in host.pop_windows():
        ptr = stack.stage_window(w)
        back = np.zeros_like(w)
        C.memory.memcpy_d2h(back, ptr, back.nbytes)
        assert np.array_equal(back, w)
    stack.close()


def test_reserved_windowed_stack():
    from trtlab_amd.core.windowed import ReservedWindowedStack

    seen = []
    st = ReservedWindowedStack(8, 3, on_window=lambda w, i: seen.append(
        (i, w.copy())))
    v = st.reserve_window()
    assert v.shape == (5,)  # stride = window - overlap
    v[:] = np.arange(5)
    st.commit_window()
    v2 = st.reserve_window()
    v2[:] = np.arange(5) + 10
    st.commit_window()
    assert len(seen) == 2
    # window 1 carries window 0's trailing overlap samples
    np.testing.assert_array_equal(seen[1][1][:3], seen[0][1][-3:])
    np.testing.assert_array_equal(seen[1][1][3:], np.arange(5) + 10)
    with pytest.raises(RuntimeError):
        st.commit_window()  # nothing reserved


def test_windowed_task_executor():
    from trtlab_amd.core.windowed import WindowedTaskExecutor

    ex = WindowedTaskExecutor(4, 1, lambda w, i: (i, float(w.sum())),
                              workers=2)
    ex.push(np.arange(16, dtype=np.float32))
    res = ex.results()
    assert [i for i, _ in res] == list(range(len(res)))
    assert len(res) >= 4
    ex.shutdown()
