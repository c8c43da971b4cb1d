CREATE TABLE mc (svc STRING, ts TIMESTAMP TIME INDEX, msg STRING, PRIMARY KEY (svc)) WITH ('append_mode'='true');
INSERT INTO mc (svc, ts, msg) VALUES ('api', 1, 'request failed with timeout'), ('api', 2, 'request ok'), ('db', 3, 'disk error detected');
SELECT ts FROM mc WHERE matches(msg, 'timeout') ORDER BY ts;
SELECT count(*) FROM mc WHERE matches(msg, 'request');
SELECT svc FROM mc WHERE matches(msg, 'error') ORDER BY svc
