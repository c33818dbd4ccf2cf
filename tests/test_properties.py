"""Property-based tests (hypothesis) for the sharding math and config."""

import sys

import pytest

try:
    from hypothesis import given, settings
    from hypothesis import strategies as st

    HAS_HYPOTHESIS = True
except ImportError:
    HAS_HYPOTHESIS = False

if not HAS_HYPOTHESIS:  # pragma: no cover
    pytest.skip('hypothesis not installed', allow_module_level=True)

from dmlcloud_amd.config import Config
from dmlcloud_amd.data import chunk_and_shard_indices, shard_indices


@settings(max_examples=200, deadline=None)
@given(
    n=st.integers(0, 500),
    world=st.integers(1, 16),
    shuffle=st.booleans(),
    seed=st.integers(0, 10),
)
def test_shard_indices_partition(n, world, shuffle, seed):
    """With even_shards, shards partition the first n - n%world elements
    into equal disjoint sets."""
    shards = [shard_indices(n, r, world, shuffle=shuffle, even_shards=True, seed=seed) for r in range(world)]
    sizes = {len(s) for s in shards}
    assert len(sizes) == 1  # equal shards
    combined = sorted(i for s in shards for i in s)
    kept = n - n % world
    assert combined == sorted(range(n))[:kept] if not shuffle else len(combined) == kept
    assert len(set(combined)) == len(combined)  # disjoint


@settings(max_examples=200, deadline=None)
@given(n=st.integers(0, 500), world=st.integers(1, 16))
def test_shard_indices_uneven_cover(n, world):
    """Without even_shards, shards cover ALL elements disjointly."""
    shards = [shard_indices(n, r, world, even_shards=False) for r in range(world)]
    combined = sorted(i for s in shards for i in s)
    assert combined == list(range(n))


@settings(max_examples=100, deadline=None)
@given(
    n=st.integers(1, 1000),
    chunk=st.integers(1, 50),
    world=st.integers(1, 8),
    overlap=st.integers(0, 10),
)
def test_chunks_within_bounds(n, chunk, world, overlap):
    """equal_chunks windows start within the data and have uniform length."""
    for r in range(world):
        for start, end in chunk_and_shard_indices(n, chunk, r, world, chunk_overlap=overlap, equal_chunks=True):
            assert 0 <= start < n or n < chunk  # starts in-range when any chunk exists
            assert end - start == chunk + overlap


@settings(max_examples=60, deadline=None)
@given(
    shapes=st.sampled_from([(3,), (2, 5), (4, 1), (2, 3, 2)]),
    n=st.integers(1, 12),
    op=st.sampled_from(['MEAN', 'SUM', 'MIN', 'MAX']),
    seed=st.integers(0, 1000),
)
def test_reducer_matches_stack_reference(shapes, n, op, seed):
    """Property: the O(1) accumulator reducer == the reference's
    torch.stack + reduce over [0] + dims for any value sequence."""
    import torch

    from dmlcloud_amd.metrics import MetricReducer, Reduction

    torch.manual_seed(seed)
    values = [torch.randn(*shapes) for _ in range(n)]
    reduction = Reduction(op)
    r = MetricReducer(reduction=reduction)
    r.extend(values)
    out = r.reduce_locally()

    stacked = torch.stack(values)
    expected = {
        'MEAN': stacked.mean(),
        'SUM': stacked.sum(),
        'MIN': stacked.amin(),
        'MAX': stacked.amax(),
    }[op]
    torch.testing.assert_close(out.to(expected.dtype), expected, rtol=1e-5, atol=1e-5)


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(1, 8),
    dim=st.sampled_from([[0], [1], [0, 1]]),
    op=st.sampled_from(['SUM', 'MIN', 'MAX', 'MEAN']),
    seed=st.integers(0, 100),
)
def test_reducer_partial_dims_property(n, dim, op, seed):
    import torch

    from dmlcloud_amd.metrics import MetricReducer, Reduction

    torch.manual_seed(seed)
    values = [torch.randn(3, 4) for _ in range(n)]
    r = MetricReducer(reduction=Reduction(op), dim=dim)
    r.extend(values)
    out = r.reduce_locally()

    stacked = torch.stack(values)
    dims = [0] + [d + 1 for d in dim]
    expected = {
        'MEAN': stacked.mean(dims),
        'SUM': stacked.sum(dims),
        'MIN': stacked.amin(dims),
        'MAX': stacked.amax(dims),
    }[op]
    torch.testing.assert_close(out.to(expected.dtype), expected, rtol=1e-5, atol=1e-5)


@settings(max_examples=100, deadline=None)
@given(
    data=st.dictionaries(
        st.text(st.characters(whitelist_categories=('Ll',), whitelist_characters='_'), min_size=1, max_size=8),
        st.one_of(st.integers(), st.floats(allow_nan=False), st.text(max_size=10), st.booleans()),
        max_size=6,
    )
)
def test_config_yaml_roundtrip(data, tmp_path_factory):
    cfg = Config.create(data)
    path = tmp_path_factory.mktemp('cfg') / 'c.yaml'
    cfg.save(path)
    assert Config.load(path) == cfg


@settings(max_examples=60, deadline=None)
@given(
    num_batches=st.integers(1, 4),
    slice_size=st.integers(1, 6),
    cols=st.integers(1, 5),
    seed=st.integers(0, 100),
)
def test_interleave_is_content_preserving(num_batches, slice_size, cols, seed):
    """Every input element appears exactly once in the interleaved output,
    and batch shapes are preserved."""
    import torch

    from dmlcloud_amd.data import interleave_batches

    torch.manual_seed(seed)
    batch_rows = slice_size * num_batches
    batches = [torch.randn(batch_rows, cols) for _ in range(num_batches)]
    out = [b.clone() for b in interleave_batches(iter(batches), num_batches)]
    assert len(out) == num_batches
    assert all(o.shape == batches[0].shape for o in out)
    source = torch.cat(batches).flatten().sort().values
    produced = torch.cat(out).flatten().sort().values
    torch.testing.assert_close(source, produced)


@settings(max_examples=80, deadline=None)
@given(
    key=st.text(st.characters(whitelist_categories=('Ll',)), min_size=1, max_size=6),
    value=st.one_of(st.integers(), st.floats(allow_nan=False, allow_infinity=False), st.booleans()),
)
def test_config_interpolation_resolves_to_referenced_value(key, value):
    """Whole-string ${key} interpolation yields the referenced value with
    its type preserved."""
    cfg = Config.create({key: value, 'alias': '${' + key + '}'})
    resolved = cfg.to_container(resolve=True)
    assert resolved['alias'] == value
    assert type(resolved['alias']) is type(value)


def _state_strategy():
    """Nested state objects shaped like real checkpoints: dicts/lists of
    tensors (varied dtypes/shapes incl. 0-dim and empty) and scalars."""
    import torch

    def tensor_strategy():
        dtypes = st.sampled_from([torch.float32, torch.float64, torch.bfloat16, torch.int64, torch.uint8, torch.bool])
        shapes = st.sampled_from([(), (1,), (3,), (2, 3), (0,), (4, 1, 2)])
        seeds = st.integers(0, 99)

        def make(args):
            dtype, shape, seed = args
            g = torch.Generator().manual_seed(seed)
            if dtype is torch.bool:
                return torch.rand(shape, generator=g) > 0.5
            if dtype in (torch.int64, torch.uint8):
                return torch.randint(0, 100, shape, generator=g, dtype=dtype)
            return torch.randn(shape, generator=g).to(dtype)

        return st.tuples(dtypes, shapes, seeds).map(make)

    leaves = st.one_of(tensor_strategy(), st.integers(), st.floats(allow_nan=False), st.text(max_size=5), st.none())
    return st.recursive(
        leaves,
        lambda children: st.one_of(
            st.dictionaries(st.text(max_size=4), children, max_size=3),
            st.lists(children, max_size=3),
        ),
        max_leaves=8,
    )


@settings(max_examples=40, deadline=None)
@given(state=_state_strategy(), seed=st.integers(0, 10))
def test_dmlt_roundtrip_property(state, seed, tmp_path_factory):
    """Any nested checkpoint-shaped object survives the .dmlt round-trip
    with identical structure, dtypes, shapes and values."""
    import torch

    from dmlcloud_amd.checkpoint import load_tensor_state, save_tensor_state

    path = tmp_path_factory.mktemp('dmlt') / 's.dmlt'
    save_tensor_state(state, path)
    loaded = load_tensor_state(path)

    def check(a, b):
        if isinstance(a, torch.Tensor):
            assert isinstance(b, torch.Tensor)
            assert a.dtype == b.dtype and tuple(a.shape) == tuple(b.shape)
            torch.testing.assert_close(b, a, equal_nan=True)
        elif isinstance(a, dict):
            assert set(a) == set(b)
            for k in a:
                check(a[k], b[k])
        elif isinstance(a, (list, tuple)):
            assert len(a) == len(b)
            for x, y in zip(a, b):
                check(x, y)
        else:
            assert a == b or (a != a and b != b)

    check(state, loaded)


@settings(max_examples=25, deadline=None)
@given(
    sizes=st.lists(
        st.tuples(st.integers(1, 13), st.integers(1, 5)),
        min_size=1,
        max_size=4,
    ),
    use_adam=st.booleans(),
    bf16=st.booleans(),
    seed=st.integers(0, 50),
)
def test_flat_replica_matches_torch_at_odd_shapes(sizes, use_adam, bf16, seed, torch_distributed_module):
    """Parameter tensors of arbitrary (non-8-aligned) sizes: the flat
    buffer's aligned views + fused optimizer == stock torch optimizer."""
    import torch

    from dmlcloud_amd.parallel import FlatAdam, FlatReplica, FlatSGD

    class ParamBag(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.ps = torch.nn.ParameterList([torch.nn.Parameter(torch.randn(*s)) for s in sizes])

        def forward(self):
            return sum((p * p).sum() for p in self.ps)

    torch.manual_seed(seed)
    m1 = ParamBag()
    m2 = ParamBag()
    m2.load_state_dict(m1.state_dict())

    dtype = torch.bfloat16 if bf16 else torch.float32
    rep = FlatReplica(m1, dtype=dtype)
    if use_adam:
        opt, ref = FlatAdam(rep, lr=1e-2), torch.optim.Adam(m2.parameters(), lr=1e-2)
    else:
        opt = FlatSGD(rep, lr=1e-2, momentum=0.9)
        ref = torch.optim.SGD(m2.parameters(), lr=1e-2, momentum=0.9)

    for _ in range(2):
        rep.zero_grad()
        m1.forward().backward()
        rep.grad_sync()
        opt.step()
        ref.zero_grad()
        m2.forward().backward()
        ref.step()

    tol = 5e-2 if bf16 else 1e-5
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        import torch as _t

        _t.testing.assert_close(p1.float(), p2.float(), rtol=tol, atol=tol)


@pytest.fixture(scope='module')
def torch_distributed_module():
    """Module-scoped dummy group (hypothesis re-runs the test body many
    times; function-scoped init/destroy per example is both slow and
    unsupported inside @given)."""
    from dmlcloud_amd.parallel import deinitialize_torch_distributed, init_process_group_dummy

    init_process_group_dummy(backend='gloo')
    yield
    deinitialize_torch_distributed()


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
