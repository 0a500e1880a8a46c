"""IoT alerting: session windows + inactivity detection per device.

Run:  python examples/iot_alerts.py
(template — point ./readings at a directory of jsonlines sensor events
{"device": "...", "ts": <unix>, "value": <float>})
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pathway_amd as pw


class Reading(pw.Schema):
    device: str
    ts: int
    value: float

readings = pw.io.jsonlines.read("./readings", schema=Reading, mode="streaming")

# bursts of activity per device: session windows with a 30s max gap
bursts = readings.windowby(
    readings.ts,
    window=pw.temporal.session(max_gap=30),
    instance=readings.device,
).reduce(
    device=pw.this._pw_instance,
    start=pw.this._pw_window_start,
    end=pw.this._pw_window_end,
    n=pw.reducers.count(),
    peak=pw.reducers.max(pw.this.value),
)

# alert when a device goes quiet for more than 2 minutes
import datetime

silence = readings.inactivity_detection(
    allowed_inactivity_period=datetime.timedelta(minutes=2),
    instance=readings.device,
)

pw.io.csv.write(bursts, "bursts.csv")
pw.io.csv.write(silence, "silence_alerts.csv")
pw.run()
