"""Model zoo for benchmarks and tests (bf16, MI355X-first)."""
