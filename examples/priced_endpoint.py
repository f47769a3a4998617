"""Endpoint with a per-task pricing policy; usage lands in /api/v1/usage
and the prometheus task_cost_cents counter."""

import beta9_amd


def handler(x: int = 1):
    return {"doubled": x * 2}


ep = beta9_amd.Endpoint(
    handler,
    cpu=1000,
    memory=2048,
    pricing=beta9_amd.PricingPolicy(cost_model="task", cost_per_task=0.002),
)

if __name__ == "__main__":
    print(ep.deploy("priced"))
    print(ep.remote(21))
