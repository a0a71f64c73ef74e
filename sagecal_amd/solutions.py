"""Solution-file I/O, byte-compatible with the reference format.

Format (writer fullbatch_mode.cpp:285-289,596-605):
  - '#' comment lines;
  - first non-comment line: freq(MHz) bandwidth(MHz) time_interval(min)
    stations clusters effective_clusters;
  - per solution interval, 8N rows: counter column then one column per
    effective cluster chunk, clusters written in REVERSE order
    (for ci=M-1..0), chunks in forward order (fullbatch_mode.cpp:598).
  - per station, the 8 values S0..S7 are the ROW-major 2x2 Jones with
    interleaved re/im: J = [S0+jS1, S2+jS3; S4+jS5, S6+jS7].

NOTE: README.md §6 describes the per-station order as COLUMN-major
(J=[S0+jS1, S4+jS5; S2+jS3, S6+jS7]) — that contradicts the code. The
writer dumps the in-memory p[] directly (fullbatch_mode.cpp:600), and
every consumer of p[] builds the Jones ROW-major (residual.c:92-99 G1
with the row-major amb() product :33). Oracle-verified
(tests/test_reference_oracle.py::test_residuals_multifreq_matches_
reference); we follow the implementation, not the README.
"""
import numpy as np
import torch

# reference per-station order == our internal row-major interleaved
# order (identity permutation; kept for documentation/symmetry)
_REF_TO_INT = np.array([0, 1, 2, 3, 4, 5, 6, 7])


def jones_to_ref_vec(J):
    """J: [..., N, 2, 2] complex -> [..., N*8] float64 in reference per
    station order."""
    v = torch.view_as_real(J).reshape(*J.shape[:-2], 8)
    return v[..., _REF_TO_INT].reshape(*J.shape[:-3], -1)


def ref_vec_to_jones(vec, N):
    """[.., N*8] reference-order reals -> [.., N, 2, 2] complex."""
    v = torch.as_tensor(vec).reshape(-1, N, 8)
    inv = np.argsort(_REF_TO_INT)
    v = v[..., inv].contiguous()
    return torch.view_as_complex(v.reshape(-1, N, 2, 2, 2).contiguous())


class SolutionWriter:
    def __init__(self, path, freq0, bandwidth, tile_minutes, N, M, Mt):
        self.f = open(path, 'w')
        self.N = N
        self.f.write("# solution file created by SAGECal\n")
        self.f.write("# freq(MHz) bandwidth(MHz) time_interval(min) stations"
                     " clusters effective_clusters\n")
        self.f.write("%lf %lf %lf %d %d %d\n" % (
            freq0 * 1e-6, bandwidth * 1e-6, tile_minutes, N, M, Mt))

    def write_tile(self, state):
        """state: CalState with J [Mt, N, 2, 2]; clusters in reverse order,
        chunks forward (fullbatch_mode.cpp:598)."""
        N = self.N
        cols = []
        M = state.M
        for ci in range(M - 1, -1, -1):
            o = state.chunk_off[ci]
            for ck in range(state.nchunks[ci]):
                cols.append(jones_to_ref_vec(
                    state.J[o + ck].cpu().to(torch.complex128)).numpy())
        cols = np.stack(cols, axis=1)  # [8N, Mt]
        for cj in range(8 * N):
            self.f.write("%d " % cj)
            self.f.write(''.join(" %e" % val for val in cols[cj]))
            self.f.write("\n")
        self.f.flush()

    def close(self):
        self.f.close()


def read_solutions(path):
    """Parse a solution file (reference read_solutions, readsky.c API
    Dirac_radio.h:102-110). Returns (header dict, list of tiles; each tile
    is a [Mt, N, 2, 2] complex tensor with columns mapped back to forward
    cluster order NOT applied — caller maps via cluster chunk counts)."""
    header = None
    rows = []
    tiles = []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith('#'):
                continue
            toks = line.split()
            if header is None:
                header = {
                    'freq_mhz': float(toks[0]), 'bw_mhz': float(toks[1]),
                    'interval_min': float(toks[2]), 'N': int(toks[3]),
                    'M': int(toks[4]), 'Mt': int(toks[5])}
                continue
            rows.append([float(t) for t in toks[1:]])
            if len(rows) == 8 * header['N']:
                arr = np.asarray(rows)  # [8N, Mt]
                rows = []
                Mt = arr.shape[1]
                cols = []
                for k in range(Mt):
                    cols.append(ref_vec_to_jones(
                        torch.tensor(arr[:, k]), header['N']))
                tiles.append(torch.cat(cols, dim=0))
    return header, tiles


def read_global_z(path):
    """Parse a global-Z consensus solution file (GlobalZWriter format =
    sagecal_master.cpp:513-517,1165-1174). Returns (header dict, list of
    tiles; each tile [Mt, Npoly, N, 2, 2] complex, columns mapped from
    file reverse order back to forward effective-cluster order)."""
    header = None
    rows = []
    tiles = []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith('#'):
                continue
            toks = line.split()
            if header is None:
                header = {
                    'freq_mhz': float(toks[0]), 'Npoly': int(toks[1]),
                    'N': int(toks[2]), 'M': int(toks[3]),
                    'Mt': int(toks[4])}
                continue
            rows.append([float(t) for t in toks[1:]])
            if len(rows) == 8 * header['N'] * header['Npoly']:
                arr = np.asarray(rows)      # [8N*Npoly, Mt] reverse order
                rows = []
                Mt = arr.shape[1]
                N, P = header['N'], header['Npoly']
                Z = torch.empty(Mt, P, N, 2, 2, dtype=torch.complex128)
                for k in range(Mt):
                    ci = Mt - 1 - k         # undo reverse cluster order
                    for p in range(P):
                        Z[ci, p] = ref_vec_to_jones(
                            torch.tensor(arr[p * 8 * N:(p + 1) * 8 * N, k]),
                            N)
                tiles.append(Z)
    return header, tiles


def reorder_read_tile(tile_J, nchunks):
    """Columns in the file are reverse-cluster-order; map a read tile
    [Mt, N, 2, 2] back to forward cluster order given per-cluster chunk
    counts."""
    M = len(nchunks)
    # file order: cluster M-1..0, chunks forward
    out = []
    pos = 0
    file_spans = {}
    for ci in range(M - 1, -1, -1):
        file_spans[ci] = (pos, pos + nchunks[ci])
        pos += nchunks[ci]
    for ci in range(M):
        s, e = file_spans[ci]
        out.append(tile_J[s:e])
    return torch.cat(out, dim=0)


class GlobalZWriter:
    """Global consensus-polynomial solution file (the MPI master's Z
    write, sagecal_master.cpp:513-517 header + :1165-1174 body): same
    layout as the J solution file but with Npoly times more rows — row
    index p in [0, 8N*Npoly), one column per EFFECTIVE cluster (Mt,
    hybrid chunks expanded — iodata.M on the master) in REVERSE order.
    Header: freq0(MHz) Npoly N Mo Mt."""

    def __init__(self, path, freq0, N, Mo, Mt, Npoly):
        self.f = open(path, 'w')
        self.N, self.M, self.Npoly = N, Mt, Npoly
        self.f.write("# solution file (Z) created by SAGECal\n")
        self.f.write("# reference_freq(MHz) polynomial_order stations "
                     "clusters effective_clusters\n")
        self.f.write("%lf %d %d %d %d\n" % (
            freq0 * 1e-6, Npoly, N, Mo, Mt))

    def write_tile(self, Z):
        """Z: [Mt, Npoly, N, 2, 2] complex (consensus.admm.ConsensusADMM.Z
        layout, per effective cluster)."""
        M, Npoly, N = Z.shape[:3]
        cols = []
        for ci in range(M - 1, -1, -1):
            col = []
            for p in range(Npoly):
                col.append(jones_to_ref_vec(
                    Z[ci, p].cpu().to(torch.complex128)).numpy())
            cols.append(np.concatenate(col))
        cols = np.stack(cols, axis=1)          # [8N*Npoly, Mt]
        for cj in range(cols.shape[0]):
            self.f.write("%d " % cj)
            self.f.write(''.join(" %e" % val for val in cols[cj]))
            self.f.write("\n")
        self.f.flush()

    def close(self):
        self.f.close()
