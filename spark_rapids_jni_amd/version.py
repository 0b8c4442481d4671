"""Spark-platform version gates (Java API parity: Version.java:19-50,
SparkPlatformType.java, version.hpp — kernel behavior switches keyed on the
Spark distribution the plugin runs against)."""
from enum import IntEnum


class SparkPlatformType(IntEnum):
    VANILLA_SPARK = 0
    DATABRICKS = 1
    CLOUDERA = 2


class Version:
    def __init__(self, platform=SparkPlatformType.VANILLA_SPARK, major=3,
                 minor=4, patch=0):
        self.platform = SparkPlatformType(platform)
        self.major, self.minor, self.patch = major, minor, patch

    def _at_least(self, major, minor):
        return (self.major, self.minor) >= (major, minor)

    def is_vanilla_320(self) -> bool:
        return (self.platform == SparkPlatformType.VANILLA_SPARK
                and (self.major, self.minor) == (3, 2))

    def is_vanilla_330_or_later(self) -> bool:
        return (self.platform == SparkPlatformType.VANILLA_SPARK
                and self._at_least(3, 3))

    def is_databricks_14_3_or_later(self) -> bool:
        return (self.platform == SparkPlatformType.DATABRICKS
                and self._at_least(14, 3))


# the active platform version (set by the integration layer at startup)
CURRENT = Version()
