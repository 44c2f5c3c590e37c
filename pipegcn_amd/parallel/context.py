"""Process-global singletons (reference: /root/reference/helper/context.py).

The model couples to the distributed runtime only through
`ctx.buffer.update(layer, h)` — same architectural seam as the reference
(SURVEY §1 "Key architectural pattern").
"""
from pipegcn_amd.parallel.buffer import Buffer
from pipegcn_amd.parallel.reducer import Reducer
from pipegcn_amd.utils.timer import comm_timer  # noqa: F401

buffer = Buffer()
reducer = Reducer()
