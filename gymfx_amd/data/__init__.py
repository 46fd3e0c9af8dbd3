from .feed import MarketData, load_csv, write_csv, synthetic_ohlcv, uptrend_ohlcv

__all__ = ["MarketData", "load_csv", "write_csv", "synthetic_ohlcv", "uptrend_ohlcv"]
