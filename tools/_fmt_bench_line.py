import json, sys
d = json.load(sys.stdin)
c = d["config"]
print("B=%d C=%d T=%d: %.0f upd/s  %.1f us/step" % (
    c["batch_per_gpu"], c["num_classes"], c["curve_thresholds"], d["value"], d["ms_per_step"] * 1000))
