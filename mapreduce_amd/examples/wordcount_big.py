"""WordCountBig — the large-corpus variant (examples/WordCountBig parity:
same functions as WordCount pointed at a big corpus; the reference only
changes the Mongo host/db, execute_BIG_server.sh:3-10).

Here: same module as examples.wordcount with a larger default partition
count; init_args carry the file list as usual.  For the GPU tier at this
scale use bench.py / mapreduce_amd.gpu.wordcount directly.
"""

from mapreduce_amd.examples.wordcount import (RESULTS, combinerfn, finalfn,  # noqa: F401
                                              mapfn, partitionfn, reducefn,
                                              taskfn)
from mapreduce_amd.examples import wordcount as _wc

associative_reducer = True
commutative_reducer = True
idempotent_reducer = True


def init(arg):
    cfg = dict(arg or {})
    cfg.setdefault("nred", 64)
    _wc.init(cfg)
