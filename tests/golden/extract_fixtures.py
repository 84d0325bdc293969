"""Extract the reference's test-data fixtures (graph DATA, not code) for use
as committed golden inputs:

  - misc/rgg2d.metis (n=1024, m=4113): BASELINE.json config 1 input.
  - tests/endtoend/data.graph.{xadj,adjncy}: the Walshaw "data" graph (n=2851) the
    reference's end-to-end test pins its determinism/cut properties on
    (shm_endtoend_test.cc:18-24).

Run in the dev container (where /root/reference is mounted):
    python tests/golden/extract_fixtures.py
Writes: tests/golden/rgg2d.metis, tests/golden/walshaw_data.json
"""

import json
import os
import shutil

REF = "/root/reference"
HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    shutil.copy(os.path.join(REF, "misc", "rgg2d.metis"),
                os.path.join(HERE, "rgg2d.metis"))
    print("copied rgg2d.metis")

    # the end-to-end fixture is stored as CSV include files
    with open(os.path.join(REF, "tests", "endtoend", "data.graph.xadj")) as f:
        xadj = [int(tok) for tok in f.read().replace(",", " ").split()]
    with open(os.path.join(REF, "tests", "endtoend", "data.graph.adjncy")) as f:
        adjncy = [int(tok) for tok in f.read().replace(",", " ").split()]
    assert xadj[-1] == len(adjncy), (xadj[-1], len(adjncy))
    with open(os.path.join(HERE, "walshaw_data.json"), "w") as f:
        json.dump({"n": len(xadj) - 1, "xadj": xadj, "adjncy": adjncy}, f)
    print(f"walshaw graph: n={len(xadj)-1} m={len(adjncy)}")


if __name__ == "__main__":
    main()
