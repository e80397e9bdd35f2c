"""WER/ExpRate scorer vs hand-computed edit distances."""

from wap_amd.eval import edit_distance, score_files, score_pairs


def test_edit_distance_hand_cases():
    assert edit_distance([], []) == 0
    assert edit_distance(list("abc"), list("abc")) == 0
    assert edit_distance(list("abc"), list("abd")) == 1     # sub
    assert edit_distance(list("abc"), list("ab")) == 1      # del
    assert edit_distance(list("abc"), list("abcd")) == 1    # ins
    assert edit_distance(list("kitten"), list("sitting")) == 3
    assert edit_distance(["\\frac", "1", "2"], ["\\frac", "2"]) == 1
    assert edit_distance(["\\frac", "1", "2"], ["\\sqrt", "3"]) == 3


def test_score_pairs():
    pairs = [
        (list("abc"), list("abc")),   # 0 err
        (list("abc"), list("abd")),   # 1 err
        (list("abcd"), list("xyzq")), # 4 err
    ]
    r = score_pairs(pairs)
    assert r.total_errors == 5 and r.total_ref_tokens == 10
    assert abs(r.wer - 0.5) < 1e-9
    assert abs(r.exprate - 1 / 3) < 1e-9
    assert abs(r.exprate_le1 - 2 / 3) < 1e-9
    assert abs(r.exprate_le2 - 2 / 3) < 1e-9


def test_score_files_contract(tmp_path):
    ref = tmp_path / "ref.txt"
    hyp = tmp_path / "hyp.txt"
    ref.write_text("u1\ta b c\nu2\tx y\n")
    hyp.write_text("u2\tx y\nu1\ta b d\n")  # order independent, keyed by uid
    r = score_files(str(ref), str(hyp))
    assert r.total_errors == 1 and abs(r.exprate - 0.5) < 1e-9


def test_missing_hypothesis_counts_as_deletions(tmp_path):
    ref = tmp_path / "ref.txt"
    hyp = tmp_path / "hyp.txt"
    ref.write_text("u1\ta b c\n")
    hyp.write_text("")
    r = score_files(str(ref), str(hyp))
    assert r.total_errors == 3 and r.exprate == 0


def test_empty_reference_edge():
    r = score_pairs([([], ["a", "b"])])
    assert r.total_errors == 2 and r.exprate == 0.0
    r2 = score_pairs([([], [])])
    assert r2.total_errors == 0 and r2.exprate == 1.0
