

def test_cntk_model_alias_feed_fetch():
    import numpy as np
    import pandas as pd
    import torch
    from mmlspark_amd.models.image_featurizer import CNTKModel
    m = CNTKModel(module=torch.nn.Linear(4, 2),
                  feedDict={"x": "feat"}, fetchDict={"score": "y"},
                  batchSize=3)
    df = pd.DataFrame({"feat": [np.arange(4, dtype=np.float32)] * 5})
    out = m.transform(df)
    assert "score" in out.columns
    assert np.stack(out["score"].to_numpy()).shape == (5, 2)
