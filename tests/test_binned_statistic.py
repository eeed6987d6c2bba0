"""
BinnedStatistic behavior (reference nbodykit/tests/test_binned_statistic.py
style) pinned against the reference's own golden JSON fixtures
(dataset_1d.json / dataset_2d.json — the ``__dtype__/__shape__/__data__``
schema of nbodykit/utils.py:381-489).
"""
import json
import os

import numpy
import pytest
from numpy.testing import assert_allclose, assert_array_equal

from nbodykit_amd.binned_statistic import BinnedStatistic
from nbodykit_amd.utils import JSONDecoder, JSONEncoder

HERE = os.path.dirname(os.path.abspath(__file__))
FIX = os.path.join(HERE, 'golden', 'ref_fixtures')


@pytest.fixture
def ds1d():
    return BinnedStatistic.from_json(os.path.join(FIX, 'dataset_1d.json'),
                                     key='data', dims=['k'])


@pytest.fixture
def ds2d():
    return BinnedStatistic.from_json(os.path.join(FIX, 'dataset_2d.json'),
                                     key='data', dims=['k', 'mu'])


def test_load_ref_fixture_1d(ds1d):
    assert ds1d.dims == ['k']
    assert ds1d.shape == (len(ds1d.edges['k']) - 1,)
    # the 1d fixture is a multipole measurement
    assert 'power_0' in ds1d.variables and 'power_2' in ds1d.variables


def test_load_ref_fixture_2d(ds2d):
    assert ds2d.dims == ['k', 'mu']
    assert len(ds2d.shape) == 2
    assert 'power' in ds2d.variables


def test_str_getitem(ds2d):
    p = ds2d['power']
    assert p.shape == ds2d.shape
    with pytest.raises(KeyError):
        ds2d['nope']


def test_index_slicing(ds2d):
    sub = ds2d[:, 0]
    assert sub.dims == ['k']
    sub2 = ds2d[2:5]
    assert sub2.shape[0] == 3


def test_sel_squeeze(ds2d):
    k0 = ds2d.coords['k'][2]
    sub = ds2d.sel(k=k0)
    assert sub.dims == ['mu']
    sub = ds2d.sel(k=[k0])
    assert sub.shape[0] == 1
    sq = sub.squeeze('k')
    assert sq.dims == ['mu']


def test_sel_nearest(ds1d):
    val = ds1d.coords['k'][3] + 1e-5
    sub = ds1d.sel(k=[val], method='nearest')
    assert sub.shape == (1,)


def test_setitem(ds1d):
    modes = numpy.ones(ds1d.shape)
    ds1d['ones'] = modes
    assert 'ones' in ds1d.variables
    assert_array_equal(ds1d['ones'], modes)


def test_reindex_average(ds2d):
    # reindex k by factor 2, modes summed (fields_to_sum)
    old = ds2d.copy()
    factor = 2
    spacing = factor * numpy.diff(ds2d.coords['k'])[0]
    new = ds2d.reindex('k', spacing, fields_to_sum=['modes'])
    assert new.shape[0] == ds2d.shape[0] // factor
    # modes sum, power nan-means
    leftover = ds2d.shape[0] % factor
    nk = new.shape[0]
    m_old = old['modes'][:nk * factor].reshape(nk, factor, -1)
    assert_allclose(new['modes'], m_old.sum(axis=1))

    avg = ds2d.average('mu')
    assert avg.dims == ['k']


def test_take(ds2d):
    sub = ds2d.take(k=ds2d.coords['k'] > 0.05)
    assert sub.shape[1] == ds2d.shape[1]
    assert (sub.coords['k'] > 0.05).all()


def test_roundtrip(tmp_path, ds2d):
    path = str(tmp_path / 'ds.json')
    ds2d.to_json(path)
    back = BinnedStatistic.from_json(path, key='data')
    assert back.dims == ds2d.dims
    for var in ds2d.variables:
        assert_allclose(numpy.nan_to_num(back[var]),
                        numpy.nan_to_num(ds2d[var]))


def test_rename(ds1d):
    ds = ds1d.copy()
    ds.rename_variable('power_0', 'Pk')
    assert 'Pk' in ds.variables and 'power_0' not in ds.variables


def test_json_schema_compat(tmp_path):
    """Files we write use the same schema the reference reads:
    numpy arrays as __dtype__/__shape__/__data__, complex as
    __complex__."""
    data = {'arr': numpy.arange(6, dtype='c16').reshape(2, 3),
            'z': 1 + 2j}
    path = str(tmp_path / 'x.json')
    with open(path, 'w') as ff:
        json.dump(data, ff, cls=JSONEncoder)
    with open(path) as ff:
        raw = json.load(ff)
    assert raw['arr']['__dtype__'] == '<c16'
    assert raw['arr']['__shape__'] == [2, 3]
    assert raw['z'] == {'__complex__': [1.0, 2.0]}
    with open(path) as ff:
        back = json.load(ff, cls=JSONDecoder)
    assert_array_equal(back['arr'], data['arr'])
    assert back['z'] == data['z']


def test_structured_dtype_roundtrip(tmp_path):
    arr = numpy.zeros(3, dtype=[('k', 'f8'), ('power', 'c16'),
                                ('modes', 'i8')])
    arr['k'] = [1, 2, 3]
    arr['power'] = [1 + 1j, 2, 3]
    arr['modes'] = [4, 5, 6]
    path = str(tmp_path / 's.json')
    with open(path, 'w') as ff:
        json.dump({'data': arr}, ff, cls=JSONEncoder)
    with open(path) as ff:
        back = json.load(ff, cls=JSONDecoder)['data']
    assert back.dtype == arr.dtype
    assert_array_equal(back, arr)
