CREATE TABLE aq ("Host" STRING, ts TIMESTAMP TIME INDEX, "Value" DOUBLE, PRIMARY KEY ("Host"));
INSERT INTO aq ("Host", ts, "Value") VALUES ('a', 1, 1.5);
SELECT "Host" AS server, "Value" * 2 AS doubled FROM aq;
SELECT "Host", "Value" FROM aq WHERE "Value" > 1
