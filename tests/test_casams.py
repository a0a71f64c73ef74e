"""CasaMS backend contract, tested against a mock casacore.tables module
(this image ships no python-casacore; the mock validates the column
mapping — autocorrelation removal, UVW metres->seconds, DATA reshaping,
any-channel flag collapse, putcol on save — per src/MS/data.cpp:604)."""
import numpy as np
import pytest
import torch

from sagecal_amd.constants import C_LIGHT
from sagecal_amd import msdata


class _MockTable:
    def __init__(self, cols, n):
        self.cols = cols
        self._n = n
        self.flushed = False

    def nrows(self):
        return self._n

    def getcol(self, name, *a):
        return self.cols[name]

    def putcol(self, name, arr):
        self.cols[name] = np.asarray(arr)

    def colnames(self):
        return list(self.cols)

    def getcoldesc(self, name):
        return {'valueType': 'complex', 'ndim': 2}

    def addcols(self, desc):
        # desc comes from _MockTables.maketabdesc -> {name: coldesc}
        for name in desc:
            self.cols[name] = None

    def close(self):
        pass

    def flush(self):
        self.flushed = True


class _MockTables:
    """Stands in for casacore.tables: main table + subtables by ::NAME."""

    def __init__(self, N=4, T=4, F=2, seed=0):
        rng = np.random.default_rng(seed)
        pairs = [(p, q) for p in range(N) for q in range(p, N)]  # incl auto
        nb_all = len(pairs)
        a1 = np.tile([p for p, _ in pairs], T)
        a2 = np.tile([q for _, q in pairs], T)
        rows = nb_all * T
        self.main = _MockTable({
            'ANTENNA1': a1, 'ANTENNA2': a2,
            'UVW': rng.standard_normal((rows, 3)) * 100.0,
            'DATA': (rng.standard_normal((rows, F, 4))
                     + 1j * rng.standard_normal((rows, F, 4))
                     ).astype(np.complex64),
            'FLAG': np.zeros((rows, F, 4), dtype=bool),
            'EXPOSURE': np.full(rows, 2.5),
        }, rows)
        # flag one cross-corr row (any channel -> row flagged);
        # pair order is (0,0),(0,1),... so row 1 is the first cross row
        self.flag_row = 1
        self.main.cols['FLAG'][self.flag_row, 1, 2] = True
        self.sub = {
            '::ANTENNA': _MockTable({}, N),
            '::SPECTRAL_WINDOW': _MockTable(
                {'CHAN_FREQ': np.array([[140e6, 160e6]]),
                 'CHAN_WIDTH': np.array([[90e3, 90e3]])}, 1),
            '::FIELD': _MockTable(
                {'PHASE_DIR': np.array([[[0.1, 0.7]]])}, 1),
        }

    def table(self, path, **kw):
        for k, t in self.sub.items():
            if path.endswith(k):
                return t
        return self.main

    @staticmethod
    def makecoldesc(name, desc):
        return {name: desc}

    @staticmethod
    def maketabdesc(coldesc):
        return coldesc


def test_casams_mapping():
    mock = _MockTables(N=4, T=4, F=2)
    ms = msdata.CasaMS('fake.ms', tilesz=2, tables_mod=mock)
    assert ms.N == 4 and ms.Nbase == 6 and ms.Ntime == 4
    assert ms.Nchan == 2 and ms.freq0 == 150e6
    assert ms.fdelta == pytest.approx(180e3)
    assert (ms.ra0, ms.dec0) == (0.1, 0.7)
    assert ms.tdelta == 2.5
    assert ms.n_tiles() == 2
    # autocorrelations excluded, pairs p<q
    assert (ms.pairs[:, 0] < ms.pairs[:, 1]).all()
    tile = ms.load_tile(0)
    # UVW converted to seconds
    sel = mock.main.cols['ANTENNA1'] != mock.main.cols['ANTENNA2']
    uvw = mock.main.cols['UVW'][sel][:12] / C_LIGHT
    assert torch.allclose(tile.u, torch.tensor(uvw[:, 0]))
    # DATA reshaped to [F, rows, 2, 2]
    d = mock.main.cols['DATA'][sel][:12]
    assert tile.xo.shape == (2, 12, 2, 2)
    assert torch.allclose(tile.xo[1, 3],
                          torch.tensor(d[3, 1].reshape(2, 2),
                                       dtype=torch.complex128))
    # any-channel flag collapses onto the row
    flagged_cross = int(np.nonzero(sel)[0].tolist().index(mock.flag_row))
    assert bool(tile.flags[flagged_cross])
    assert int(tile.flags.sum()) == 1


def test_casams_write_roundtrip():
    mock = _MockTables(N=4, T=4, F=2)
    ms = msdata.CasaMS('fake.ms', tilesz=2, tables_mod=mock)
    tile = ms.load_tile(1)
    ms.write_column('residual', 1, tile.xo * 0.5)
    data_before = mock.main.cols['DATA'].copy()
    ms.save()
    assert mock.main.flushed
    # missing CORRECTED_DATA is CREATED (addImagingColumns analog), the
    # raw DATA column is never silently overwritten (ADVICE r1 high;
    # reference errors on a missing output column, data.cpp:1404)
    assert np.array_equal(mock.main.cols['DATA'], data_before)
    sel = mock.main.cols['ANTENNA1'] != mock.main.cols['ANTENNA2']
    got = mock.main.cols['CORRECTED_DATA'][sel][12:24]
    want = tile.xo.permute(1, 0, 2, 3).numpy().reshape(12, 2, 4) * 0.5
    assert np.allclose(got, want, atol=1e-6)
    # untouched rows were initialized from DATA
    got0 = mock.main.cols['CORRECTED_DATA'][sel][:12]
    assert np.allclose(got0, data_before[sel][:12])


def test_casams_save_raises_when_column_uncreatable():
    mock = _MockTables(N=4, T=4, F=2)
    mock.main.addcols = None  # table refuses addcols
    ms = msdata.CasaMS('fake.ms', tilesz=2, tables_mod=mock)
    tile = ms.load_tile(0)
    ms.write_column('residual', 0, tile.xo * 0.5)
    with pytest.raises(RuntimeError, match='could not be created'):
        ms.save()
    # DATA untouched even on failure
    assert 'CORRECTED_DATA' not in mock.main.cols


def test_casams_gated_without_casacore():
    with pytest.raises(RuntimeError, match='python-casacore'):
        msdata.CasaMS('fake.ms')
