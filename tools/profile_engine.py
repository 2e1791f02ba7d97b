"""cProfile the engine bench in-process; writes gpurun_out/pyprof.txt."""
import cProfile
import io
import pstats
import runpy
import sys

sys.argv = ["bench.py", "--mode", "engine", "--steps", "8", "--warmup", "2"]
pr = cProfile.Profile()
pr.enable()
try:
    runpy.run_path("bench.py", run_name="__main__")
except SystemExit:
    pass
pr.disable()
st = pstats.Stats(pr)
st.sort_stats("tottime")
buf = io.StringIO()
st.stream = buf
st.print_stats(40)
open("gpurun_out/pyprof.txt", "w").write(buf.getvalue())
