"""API group llmd.ai — VariantAutoscaling CRD types."""
