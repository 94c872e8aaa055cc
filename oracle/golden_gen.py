# oracle/golden_gen.py — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py).
#
# Generates tests/golden/*.json: the reference's own golden unit tests,
# re-derived by hand from the test SOURCE (inputs and expected outputs read
# from the cited lines — no reference code or data files are copied).
# Run:  python -m oracle.golden_gen
#
# These fixtures PIN the oracle (tests/test_oracle.py asserts
# oracle(inputs) == expected); GPU parity tests then pin the product
# against the oracle.
import json
import os

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = os.path.join(os.path.dirname(HERE), "tests", "golden")


def merge_stream_cases():
    """read.rs:512-573 test_merge_stream — one pre-sorted stream of 3
    batches, pk1 u8 + binary value + per-row __seq__ 1..9, both operators.

    input rows (pk1, value, seq):
      (11,"1",1)(11,"2",2)(12,"3",3)(12,"4",4)(13,"5",5) | (13,"6",6)(13,"7",7) | (13,"8",8)(14,"9",9)
    expected LastValue  (read.rs:514-527): (11,"2")(12,"4")(13,"8")(14,"9")
    expected BytesMerge (read.rs:529-540): (11,"12")(12,"34")(13,"5678")(14,"9")
    """
    return {
        "cite": "read.rs:512-573",
        "num_primary_keys": 1,
        "value_idxes": [1],
        "input": {
            "pk1": [11, 11, 12, 12, 13, 13, 13, 13, 14],
            "value": ["1", "2", "3", "4", "5", "6", "7", "8", "9"],
            "seq": [1, 2, 3, 4, 5, 6, 7, 8, 9],
        },
        "expected_last": {"pk1": [11, 12, 13, 14], "value": ["2", "4", "8", "9"]},
        "expected_append": {"pk1": [11, 12, 13, 14], "value": ["12", "34", "5678", "9"]},
    }


def operator_cases():
    """operator.rs:119-158 — single-group merges.
    LastValue (operator.rs:119-135): rows (11,100,[2,7,4,1]) -> (11,100,1).
    BytesMerge (operator.rs:137-158): values one,two,three,four -> concat."""
    return {
        "cite": "operator.rs:119-158",
        "last": {
            "input": {"pk1": [11, 11, 11, 11], "pk2": [100, 100, 100, 100],
                      "value": [2, 7, 4, 1]},
            "expected": {"pk1": [11], "pk2": [100], "value": [1]},
        },
        "append": {
            "input": {"pk1": [11, 11, 11, 11], "pk2": [100, 100, 100, 100],
                      "value": ["one", "two", "three", "four"]},
            "expected": {"pk1": [11], "pk2": [100], "value": ["onetwothreefour"]},
        },
    }


def storage_write_scan_case():
    """storage.rs:392-491 test_storage_write_and_scan — two writes into one
    2h segment become two SSTs (seq1 < seq2); Overwrite scan dedups (11,100)
    keeping the newer file's row; predicate pk1 == 11 filters BEFORE merge.

    write1 (seq=1): (11,100,2)(11,100,7)(9,1,4)(10,2,6)(5,3,1)  [sorted on write]
    write2 (seq=2): (11,100,22)(11,99,77)(9,1,44)(10,2,66)
    expected scan (storage.rs:448-461, stream concat):
      (5,3,1)(9,1,44)(10,2,66)(11,99,77)(11,100,22)
    expected with pk1==11 (storage.rs:475-489):
      (11,99,77)(11,100,22)
    """
    return {
        "cite": "storage.rs:392-491",
        "num_primary_keys": 2,
        "sst1": {"pk1": [5, 9, 10, 11, 11], "pk2": [3, 1, 2, 100, 100],
                 "value": [1, 4, 6, 2, 7], "seq": 1},
        "sst2": {"pk1": [9, 10, 11, 11], "pk2": [1, 2, 99, 100],
                 "value": [44, 66, 77, 22], "seq": 2},
        "expected": {"pk1": [5, 9, 10, 11, 11], "pk2": [3, 1, 2, 99, 100],
                     "value": [1, 44, 66, 77, 22]},
        "expected_pk1_eq_11": {"pk1": [11, 11], "pk2": [99, 100],
                               "value": [77, 22]},
    }


def sort_batch_case():
    """storage.rs:493-536 test_storage_sort_batch — stable sort by 1 PK."""
    return {
        "cite": "storage.rs:493-536",
        "num_primary_keys": 1,
        "input": {"a": [2, 1, 3, 4, 8, 6, 5, 7], "b": [1, 3, 4, 8, 2, 6, 5, 7],
                  "c": [8, 6, 2, 4, 3, 1, 5, 7], "d": [2, 7, 4, 6, 1, 3, 5, 8]},
        "expected": {"a": [1, 2, 3, 4, 5, 6, 7, 8], "b": [3, 1, 4, 8, 5, 6, 7, 2],
                     "c": [6, 8, 2, 4, 5, 1, 7, 3], "d": [7, 2, 4, 6, 5, 3, 8, 1]},
    }


def schema_cases():
    """types.rs:246-302 — truncate_by table and fill_required_projections
    table (schema: 2 PKs + 1 value => seq_idx=3, reserved_idx=4)."""
    return {
        "cite": "types.rs:246-302",
        "truncate_by": [[0, 20, 0], [10, 20, 0], [20, 20, 20], [30, 20, 20],
                        [40, 20, 40], [41, 20, 40]],
        "fill_required_projections": {
            "num_primary_keys": 2, "seq_idx": 3,
            "cases": [[None, None], [[], [0, 1, 3]], [[1], [1, 0, 3]],
                      [[2], [2, 0, 1, 3]]],
        },
    }


def main():
    os.makedirs(GOLDEN, exist_ok=True)
    fixtures = {
        "merge_stream.json": merge_stream_cases(),
        "operators.json": operator_cases(),
        "storage_write_scan.json": storage_write_scan_case(),
        "sort_batch.json": sort_batch_case(),
        "schema.json": schema_cases(),
    }
    for name, data in fixtures.items():
        with open(os.path.join(GOLDEN, name), "w") as f:
            json.dump(data, f, indent=1)
        print("wrote", name)


if __name__ == "__main__":
    main()
