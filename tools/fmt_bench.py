import json, sys
d = json.load(sys.stdin)
c = d["config"]
print("B=%6d d=%4d %s: %7.2f ms/step  %6.2f M pairs/s"
      % (c["global_batch"], c["emb_dim"], d["dtype"], d["ms_per_step"],
         d["value"] / 1e6))
