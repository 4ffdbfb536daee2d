"""CPU tests of the `solver` CLI (§8(f) row 3): translation global
optimisation over StitchingResults links, the stale-hash drop rule
(reference Solver.java:404-415), and the registration update side
effect."""

import os
import subprocess
import xml.etree.ElementTree as ET

import numpy as np

from tests import n5util
from tests.test_cli_host import BIN, run


def write_xml_with_links(path, entries, posB=40.0):
    n5util.make_dataset_xml(
        path, "input.n5",
        [dict(id=0, dims=(64, 64, 64), pos=(0.0, 0.0, 0.0)),
         dict(id=1, dims=(64, 64, 64), pos=(posB, 0.0, 0.0))],
    )
    text = open(path).read()
    sr = "  <StitchingResults>\n"
    for e in entries:
        sr += (
            "    <PairwiseResult>\n"
            f"      <ViewIdsA>{e['a']}</ViewIdsA>\n"
            f"      <ViewIdsB>{e['b']}</ViewIdsB>\n"
            f"      <Matrix>1 0 0 {e['ws'][0]} 0 1 0 {e['ws'][1]} "
            f"0 0 1 {e['ws'][2]}</Matrix>\n"
            "      <BoundingBoxMin>0 0 0</BoundingBoxMin>\n"
            "      <BoundingBoxMax>1 1 1</BoundingBoxMax>\n"
            f"      <Correlation>{e.get('r', 0.95)}</Correlation>\n"
            f"      <Hash>{e['hash']}</Hash>\n"
            "    </PairwiseResult>\n"
        )
    sr += "  </StitchingResults>\n"
    text = text.replace("</SpimData>", sr + "</SpimData>")
    open(path, "w").write(text)


def model_translations(path):
    """Effective (outermost-first concatenated) translation per setup."""
    tree = ET.parse(path)
    out = {}
    for vr in tree.getroot().iter("ViewRegistration"):
        setup = int(vr.get("setup"))
        m = np.eye(4)
        first = True
        for vt in vr.findall("ViewTransform"):
            a = np.fromstring(vt.find("affine").text, sep=" ").reshape(3, 4)
            m4 = np.vstack([a, [0, 0, 0, 1]])
            m = m4 if first else m @ m4
            first = False
        out[setup] = m[:3, 3]
    return out


def test_solver_translation_solve(tmp_path):
    xml = os.path.join(str(tmp_path), "dataset.xml")
    # link says B's content appears at ws = -e relative to registrations;
    # solver should move B by +e = -ws
    e = (2.5, -1.5, 1.0)
    hash_ok = 3.0 + (3.0 + 40.0)  # [PIN-HASH] sums of both 3x4 models
    write_xml_with_links(
        xml,
        [dict(a="0,0", b="0,1", ws=(-e[0], -e[1], -e[2]), hash=hash_ok)],
    )
    r = run([os.path.join(BIN, "solver"), "-x", xml])
    assert r.returncode == 0, r.stderr + r.stdout
    t = model_translations(xml)
    assert np.allclose(t[0], [0, 0, 0], atol=1e-9)  # fixed view
    assert np.allclose(t[1], [40 + e[0], e[1], e[2]], atol=1e-6), t[1]


def test_solver_drops_stale_hash(tmp_path):
    xml = os.path.join(str(tmp_path), "dataset.xml")
    write_xml_with_links(
        xml, [dict(a="0,0", b="0,1", ws=(-2.0, 0.0, 0.0), hash=123.456)]
    )
    r = run([os.path.join(BIN, "solver"), "-x", xml])
    assert r.returncode == 1  # nothing to solve after the stale drop
    assert "1 stale-hash" in r.stdout


def test_solver_drops_low_r(tmp_path):
    xml = os.path.join(str(tmp_path), "dataset.xml")
    hash_ok = 3.0 + 43.0
    write_xml_with_links(
        xml,
        [dict(a="0,0", b="0,1", ws=(-2.0, 0.0, 0.0), hash=hash_ok, r=0.1)],
    )
    r = run([os.path.join(BIN, "solver"), "-x", xml])
    assert r.returncode == 1
    assert "1 below minR" in r.stdout


def test_solver_three_view_chain(tmp_path):
    """0-1 and 1-2 links chain; adjustments accumulate."""
    xml = os.path.join(str(tmp_path), "dataset.xml")
    n5util.make_dataset_xml(
        xml, "input.n5",
        [dict(id=0, dims=(64, 64, 64), pos=(0.0, 0.0, 0.0)),
         dict(id=1, dims=(64, 64, 64), pos=(40.0, 0.0, 0.0)),
         dict(id=2, dims=(64, 64, 64), pos=(80.0, 0.0, 0.0))],
    )
    h01 = 3.0 + 43.0
    h12 = 43.0 + 83.0
    text = open(xml).read()
    sr = "  <StitchingResults>\n"
    for a, b, ws, h in [("0,0", "0,1", -1.0, h01),
                        ("0,1", "0,2", -2.0, h12)]:
        sr += (
            "    <PairwiseResult>\n"
            f"      <ViewIdsA>{a}</ViewIdsA>\n      <ViewIdsB>{b}</ViewIdsB>\n"
            f"      <Matrix>1 0 0 {ws} 0 1 0 0 0 0 1 0</Matrix>\n"
            "      <BoundingBoxMin>0 0 0</BoundingBoxMin>\n"
            "      <BoundingBoxMax>1 1 1</BoundingBoxMax>\n"
            "      <Correlation>0.9</Correlation>\n"
            f"      <Hash>{h}</Hash>\n    </PairwiseResult>\n"
        )
    sr += "  </StitchingResults>\n"
    open(xml, "w").write(text.replace("</SpimData>", sr + "</SpimData>"))
    r = run([os.path.join(BIN, "solver"), "-x", xml])
    assert r.returncode == 0, r.stderr + r.stdout
    t = model_translations(xml)
    assert np.allclose(t[1], [41.0, 0, 0], atol=1e-6)
    assert np.allclose(t[2], [83.0, 0, 0], atol=1e-6)


def test_solver_isolated_view_unchanged(tmp_path):
    """A view with no (accepted) links must keep its registration and
    the solver must not fail on the disconnected graph (the reference
    optimises per connected component; an isolated tile stays put)."""
    xml = os.path.join(str(tmp_path), "d.xml")
    n5util.make_dataset_xml(
        xml, "input.n5",
        [dict(id=0, dims=(64, 64, 64), pos=(0.0, 0.0, 0.0)),
         dict(id=1, dims=(64, 64, 64), pos=(40.0, 0.0, 0.0)),
         dict(id=2, dims=(64, 64, 64), pos=(0.0, 40.0, 0.0))],
    )
    # link only 0<->1; setup 2 is isolated
    text = open(xml).read()
    sr = (
        "  <StitchingResults>\n"
        "    <PairwiseResult>\n"
        "      <ViewIdsA>0,0</ViewIdsA>\n"
        "      <ViewIdsB>0,1</ViewIdsB>\n"
        "      <Matrix>1 0 0 -1.5 0 1 0 0 0 0 1 0</Matrix>\n"
        "      <BoundingBoxMin>0 0 0</BoundingBoxMin>\n"
        "      <BoundingBoxMax>1 1 1</BoundingBoxMax>\n"
        "      <Correlation>0.95</Correlation>\n"
        f"      <Hash>{3.0 + (3.0 + 40.0)}</Hash>\n"  # [PIN-HASH]
        "    </PairwiseResult>\n"
        "  </StitchingResults>\n"
    )
    open(xml, "w").write(text.replace("</SpimData>", sr + "</SpimData>"))
    before = model_translations(xml)
    r = run([os.path.join(BIN, "solver"), "-x", xml])
    assert r.returncode == 0, r.stderr + r.stdout
    after = model_translations(xml)
    # isolated view 2: unchanged
    assert np.allclose(after[2], before[2])
    # linked pair: ws = -e means B's content sits at -1.5 relative to
    # the registrations; solver moves B by +e = -ws (the convention
    # pinned by test_solver_translation_solve)
    rel = after[1][0] - after[0][0]
    assert abs(rel - (40.0 + 1.5)) < 1e-6


def test_solver_malformed_xml_clean_error(tmp_path):
    """Truncated/invalid XML must produce a clean nonzero exit, not a
    crash (the host parsers are exercised by every CLI)."""
    xml = os.path.join(str(tmp_path), "broken.xml")
    open(xml, "w").write("<SpimData version=\"0.2\"><SequenceDescription>")
    r = run([os.path.join(BIN, "solver"), "-x", xml])
    assert r.returncode != 0
    assert r.returncode < 128, "must exit, not die on a signal"


def test_solver_missing_file_clean_error(tmp_path):
    r = run([os.path.join(BIN, "solver"), "-x",
             os.path.join(str(tmp_path), "nope.xml")])
    assert r.returncode != 0 and r.returncode < 128


def test_solver_correlation_weighted(tmp_path):
    """[PIN-WEIGHT] w = max(0, r): an inconsistent low-correlation cycle
    link pulls the solution by exactly the weighted-Laplacian amount
    (checked against an independent numpy solve), and NOT by the
    unweighted amount."""
    xml = os.path.join(str(tmp_path), "dataset.xml")
    n5util.make_dataset_xml(
        xml, "input.n5",
        [dict(id=0, dims=(64, 64, 64), pos=(0.0, 0.0, 0.0)),
         dict(id=1, dims=(64, 64, 64), pos=(40.0, 0.0, 0.0)),
         dict(id=2, dims=(64, 64, 64), pos=(80.0, 0.0, 0.0))],
    )
    # links as (a, b, ws_x, r, hash = 3+posA + 3+posB of the current
    # identity+translation registrations)
    links = [("0,0", "0,1", -1.0, 0.9, 3.0 + 43.0),
             ("0,1", "0,2", -1.0, 0.9, 43.0 + 83.0),
             ("0,0", "0,2", -5.0, 0.35, 3.0 + 83.0)]  # above minR=0.3
    text = open(xml).read()
    sr = "  <StitchingResults>\n"
    for a, b, ws, r_, h in links:
        sr += (
            "    <PairwiseResult>\n"
            f"      <ViewIdsA>{a}</ViewIdsA>\n      <ViewIdsB>{b}</ViewIdsB>\n"
            f"      <Matrix>1 0 0 {ws} 0 1 0 0 0 0 1 0</Matrix>\n"
            "      <BoundingBoxMin>0 0 0</BoundingBoxMin>\n"
            "      <BoundingBoxMax>1 1 1</BoundingBoxMax>\n"
            f"      <Correlation>{r_}</Correlation>\n"
            f"      <Hash>{h}</Hash>\n    </PairwiseResult>\n"
        )
    sr += "  </StitchingResults>\n"
    open(xml, "w").write(text.replace("</SpimData>", sr + "</SpimData>"))
    r = run([os.path.join(BIN, "solver"), "-x", xml])
    assert r.returncode == 0, r.stderr + r.stdout
    t = model_translations(xml)

    def solve(ws_list):
        # minimize sum w (d_b - d_a - m)^2, d_0 fixed at 0; m = -ws
        A = np.zeros((2, 2))
        rhs = np.zeros(2)
        for (ia, ib, ws, w) in ws_list:
            m = -ws
            for (i, s) in ((ia, -1.0), (ib, 1.0)):
                if i == 0:
                    continue
                A[i - 1, i - 1] += w
                other = ib if i == ia else ia
                if other != 0:
                    A[i - 1, other - 1] -= w
                rhs[i - 1] += s * w * m
        return np.linalg.solve(A, rhs)

    lw = [(0, 1, -1.0, 0.9), (1, 2, -1.0, 0.9), (0, 2, -5.0, 0.35)]
    dw = solve(lw)
    du = solve([(a, b, ws, 1.0) for (a, b, ws, _w) in lw])
    assert abs(dw[0] - du[0]) > 0.1  # the weight matters for this graph
    assert np.allclose(t[1], [40.0 + dw[0], 0, 0], atol=1e-9)
    assert np.allclose(t[2], [80.0 + dw[1], 0, 0], atol=1e-9)


def test_solver_fixed_views_flag(tmp_path):
    """--fixedViews tp,setup: anchoring view 1 instead of the default
    first view moves view 0 by the opposite amount."""
    xml = os.path.join(str(tmp_path), "dataset.xml")
    e = (2.5, -1.5, 1.0)
    hash_ok = 3.0 + (3.0 + 40.0)
    write_xml_with_links(
        xml,
        [dict(a="0,0", b="0,1", ws=(-e[0], -e[1], -e[2]), hash=hash_ok)],
    )
    r = run([os.path.join(BIN, "solver"), "-x", xml,
             "--fixedViews", "0,1"])
    assert r.returncode == 0, r.stderr + r.stdout
    t = model_translations(xml)
    assert np.allclose(t[1], [40, 0, 0], atol=1e-9)  # anchored
    assert np.allclose(t[0], [-e[0], -e[1], -e[2]], atol=1e-6), t[0]
