CREATE TABLE pr (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h)) PARTITION ON COLUMNS (h) (h < 'g', h >= 'g' AND h < 'p', h >= 'p');
INSERT INTO pr (h, ts, v) VALUES ('alpha',1,1.0),('golf',2,2.0),('papa',3,3.0),('zulu',4,4.0);
SELECT count(*) FROM pr;
SELECT h, v FROM pr WHERE h = 'golf';
SELECT h FROM pr WHERE h >= 'p' ORDER BY h;
SELECT min(v) AS lo, max(v) AS hi FROM pr;
SHOW CREATE TABLE pr
