"""SAR time-format/startTime params (SAR.scala:50-53,240-244)."""
import numpy as np
import pandas as pd

from mmlspark_amd.models.sar import SAR


def test_sar_string_timestamps_and_start_time():
    """String timeCol parses; startTime shifts the decay reference
    (SAR.scala startTime/activityTimeFormat)."""
    df = pd.DataFrame({
        "userIdx": [0, 0, 1, 1, 0, 1] * 4,
        "itemIdx": [0, 1, 0, 2, 1, 2] * 4,
        "rating": [1.0] * 24,
        "ts": ["2026/01/01T00:00:00", "2026/03/01T00:00:00"] * 12,
    })
    m = SAR(timeCol="ts", timeDecayCoeff=30,
            startTime="2026-06-01").fit(df)
    A = m.get("sarArrays")["affinity"]
    assert A.shape[0] == 2 and np.isfinite(A).all() and (A >= 0).all()
    # older events decayed more: the Jan interactions contribute less
    m2 = SAR(timeCol="ts", timeDecayCoeff=30).fit(df)
    A2 = m2.get("sarArrays")["affinity"]
    assert not np.allclose(A, A2)
    idf = m.getItemDataFrame()
    udf_ = m.getUserDataFrame()
    assert len(idf) == 3 and len(udf_) == 2
    assert np.asarray(idf["similarity"].iloc[0]).shape == (3,)
